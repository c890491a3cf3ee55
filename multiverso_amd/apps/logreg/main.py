"""CLI: python -m multiverso_amd.apps.logreg.main <config_file>
(reference Applications/LogisticRegression/src/main.cpp:7-13)."""

import sys

import multiverso_amd as mv

from .logreg import LogReg


def main() -> None:
    if len(sys.argv) < 2:
        print("usage: python -m multiverso_amd.apps.logreg.main <config>")
        sys.exit(1)
    lr = LogReg(sys.argv[1])
    cfg = lr.cfg
    if cfg.train_file:
        lr.train()
    if cfg.output_model_file:
        lr.save_model()
    if cfg.test_file:
        lr.test(output_file=cfg.output_file)
    if cfg.use_ps:
        mv.shutdown()


if __name__ == "__main__":
    main()
