"""Embedding serving on the parameter server — the deployment story for
a trained WordEmbedding model: the embedding matrix lives in a
row-sharded MatrixTable (HBM-resident on GPU nodes; one process per GPU
can shard multi-hundred-GB tables), and an HTTP front end serves vector
lookups and nearest-neighbor queries straight off the table's rows.

The reference stops at saving word2vec files; this example closes the
loop to "production serving" with the same table primitives.

  # 1) train (examples/run_wordembedding.sh) -> embedding.bin
  # 2) serve:
  python examples/serve_embeddings.py embedding.bin --port 8080
  # 3) query:
  curl localhost:8080/vec/king
  curl localhost:8080/nn/king?k=5
"""

import argparse
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def load_word2vec(path: str):
    """Read word2vec text or binary format (the app's save_embedding
    output / distributed_wordembedding.cpp:263-306)."""
    words, vecs = [], []
    with open(path, "rb") as f:
        header = f.readline().split()
        vocab, dim = int(header[0]), int(header[1])
        blob = f.read()
    # binary: "word " + dim*4 bytes + "\n"; text: "word f f f ...\n".
    # Raw float bytes virtually always contain control bytes (<0x09);
    # text never does.
    probe = blob[:256]
    is_binary = any(b < 9 for b in probe)
    off = 0
    if is_binary:
        for _ in range(vocab):
            sp = blob.index(b" ", off)
            words.append(blob[off:sp].decode())
            vec = struct.unpack(f"<{dim}f", blob[sp + 1:sp + 1 + dim * 4])
            vecs.append(vec)
            off = sp + 1 + dim * 4
            if off < len(blob) and blob[off:off + 1] == b"\n":
                off += 1
    else:
        for line in blob.decode().strip().split("\n"):
            parts = line.split()
            words.append(parts[0])
            vecs.append([float(x) for x in parts[1:]])
    return words, torch.tensor(vecs, dtype=torch.float32)


def build_app(table, words, dim):
    from fastapi import FastAPI, HTTPException
    app = FastAPI(title="multiverso_amd embedding server")
    wid = {w: i for i, w in enumerate(words)}

    def vec_of(word: str) -> torch.Tensor:
        if word not in wid:
            raise HTTPException(404, f"unknown word '{word}'")
        return table.get_rows([wid[word]])[0]

    @app.get("/healthz")
    def healthz():
        return {"vocab": len(words), "dim": dim,
                "device": str(table.device)}

    @app.get("/vec/{word}")
    def vec(word: str):
        return {"word": word, "vector": vec_of(word).cpu().tolist()}

    @app.get("/nn/{word}")
    def nn(word: str, k: int = 5):
        q = vec_of(word)
        full = table.get()                      # all-gather of shards
        sims = torch.nn.functional.cosine_similarity(
            full, q.unsqueeze(0), dim=1)
        top = torch.topk(sims, min(k + 1, len(words)))
        out = [{"word": words[int(i)], "score": float(s)}
               for s, i in zip(top.values, top.indices)
               if words[int(i)] != word][:k]
        return {"word": word, "neighbors": out}

    return app


def main():
    p = argparse.ArgumentParser()
    p.add_argument("embedding_file")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--host", default="127.0.0.1")
    args = p.parse_args()

    import multiverso_amd as mv
    mv.init()
    words, vecs = load_word2vec(args.embedding_file)
    table = mv.MatrixTable(len(words), vecs.shape[1])
    table.add(vecs)     # master loads; at N>1 use the handler protocol
    table.flush()
    print(f"serving {len(words)} x {vecs.shape[1]} embeddings from "
          f"{table.device}", flush=True)

    import uvicorn
    uvicorn.run(build_app(table, words, vecs.shape[1]),
                host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
