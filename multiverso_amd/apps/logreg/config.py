"""LogisticRegression config — key=value file parsing.

Capability parity with the reference config system
(Applications/LogisticRegression/src/configure.h:20-97,
configure.cpp:32-84): same key names and defaults for the documented keys;
unknown keys warn and are ignored."""

from __future__ import annotations

from dataclasses import dataclass, fields


@dataclass
class LogRegConfig:
    input_size: int = 0
    output_size: int = 1
    train_epoch: int = 1
    minibatch_size: int = 20
    learning_rate: float = 0.1
    learning_rate_coef: float = 1.0
    regular_type: str = "none"          # none | l1 | l2
    regular_coef: float = 0.0001
    objective_type: str = "sigmoid"     # sigmoid | softmax | ftrl
    updater_type: str = "sgd"           # default | sgd | ftrl | adagrad
    # reference configure.h:24-25 flag: True = libsvm "label k:v ..."
    # text + sparse model file; False = DENSE "label v v ..." text (the
    # reference's own mnist.config runs dense softmax) + dense model
    # file. Default True here (sparse is this framework's primary
    # path); the reference defaulted false — config files state it
    # explicitly either way, as the reference's example does.
    sparse: bool = True
    use_ps: bool = False
    # reference default true (configure.h:84): its effect — overlapping
    # the next chunk's model pull and data parse with training — is
    # structural here: the SampleReader parses ahead on its own thread
    # and single-rank pulls are stream-ordered behind the training
    # kernels (no host sync), so the flag is accepted for config parity
    # and changes nothing.
    pipeline: bool = True
    sync_frequency: int = 1
    reader_type: str = "default"        # default | weight | bsparse
    train_file: str = ""
    test_file: str = ""
    output_model_file: str = ""
    init_model_file: str = ""
    output_file: str = "result.txt"
    read_buffer_size: int = 100_000
    show_time_per_sample: int = 1_000_000
    # FTRL hyperparameters (objective.cpp:250-258)
    alpha: float = 0.1
    beta: float = 1.0
    lambda1: float = 0.01
    lambda2: float = 0.01

    @classmethod
    def from_file(cls, path: str) -> "LogRegConfig":
        cfg = cls()
        types = {f.name: f.type for f in fields(cls)}
        with open(path) as f:
            for line in f:
                line = line.split("#")[0].strip()
                if not line or "=" not in line:
                    continue
                key, _, val = line.partition("=")
                key, val = key.strip(), val.strip()
                if not hasattr(cfg, key):
                    from multiverso_amd.log import log
                    log.error(f"[logreg] unknown config key '{key}' ignored")
                    continue
                cur = getattr(cfg, key)
                if isinstance(cur, bool):
                    setattr(cfg, key, val.lower() in ("1", "true", "yes"))
                elif isinstance(cur, int):
                    setattr(cfg, key, int(float(val)))
                elif isinstance(cur, float):
                    setattr(cfg, key, float(val))
                else:
                    setattr(cfg, key, val)
        return cfg
