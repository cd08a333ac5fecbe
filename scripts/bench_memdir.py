"""BASELINE config[3]: memdir semantic search — bge-base-class embeddings
on MFMA, 1M-memory synthetic corpus, 1 GPU.

Measures:
  1. real encode throughput (texts/s) through the GPU encoder
     (fei_amd/models/bge.py: GEMMs on hipBLASLt/MFMA + HIP attention/norm
     kernels) on synthetic texts
  2. query latency/QPS at 1M corpus: encode query + cosine GEMV over
     [1M, 768] + top-k. Corpus embeddings are synthetic (random unit
     vectors) — encoding 1M texts for real would only scale measurement 1,
     and the QUERY path (the serving metric) is identical.
  3. the reference-style lexical full scan for contrast (in-memory scan of
     1M synthetic docs — the reference re-reads files, so this flatters it)

Prints one JSON line per measurement.
"""

import json
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def main():
    has_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0" if has_gpu else "cpu")
    N = int(os.environ.get("CORPUS_N", "1000000" if has_gpu else "20000"))
    DIM = 768

    from fei_amd.models.bge import BgeEncoder
    enc = BgeEncoder(device=dev)

    # 1) encode throughput on real synthetic texts
    rng = random.Random(0)
    words = ["kernel", "memory", "agent", "rocm", "tile", "cache", "tensor",
             "search", "graph", "token", "wave", "stream", "note", "task"]
    texts = [" ".join(rng.choices(words, k=rng.randint(8, 40)))
             for _ in range(2048 if has_gpu else 64)]
    if has_gpu:
        enc.encode_texts(texts[:64])            # warm
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    emb = enc.encode_texts(texts, batch_size=128)
    if has_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({"metric": "memdir encode throughput", "unit": "texts/s",
                      "value": round(len(texts) / dt, 1),
                      "n_texts": len(texts), "dim": DIM,
                      "device": str(dev), "data": "synthetic"}))

    # 2) query latency at N-corpus
    g = torch.Generator(device=dev).manual_seed(1)
    corpus = torch.randn(N, DIM, generator=g, device=dev)
    corpus = F.normalize(corpus, dim=-1)
    queries = [" ".join(rng.choices(words, k=12)) for _ in range(100)]

    def one_query(q):
        qe = enc.encode_texts([q])[0]
        scores = corpus @ qe
        return torch.topk(scores, 10)

    one_query(queries[0])                        # warm
    if has_gpu:
        torch.cuda.synchronize()
    lat = []
    for q in queries:
        t0 = time.perf_counter()
        one_query(q)
        if has_gpu:
            torch.cuda.synchronize()
        lat.append(time.perf_counter() - t0)
    lat.sort()
    p50 = lat[len(lat) // 2] * 1000
    p95 = lat[int(len(lat) * 0.95)] * 1000
    print(json.dumps({"metric": "memdir semantic query (encode+GEMV+topk)",
                      "unit": "ms", "p50": round(p50, 2), "p95": round(p95, 2),
                      "qps": round(1000 / p50, 1), "corpus": N, "topk": 10,
                      "device": str(dev), "data": "synthetic"}))

    # 3) reference-style lexical scan for contrast (in-memory; the reference
    #    re-reads every FILE per query: memdir_tools/search.py:361-367)
    docs = [" ".join(rng.choices(words, k=20)) for _ in range(min(N, 200000))]
    t0 = time.perf_counter()
    hits = sum(1 for d in docs if "kernel" in d)
    dt = time.perf_counter() - t0
    rate = len(docs) / dt
    print(json.dumps({"metric": "lexical full-scan (reference style)",
                      "unit": "docs/s", "value": round(rate),
                      "scanned": len(docs), "hits": hits,
                      "projected_ms_for_corpus": round(N / rate * 1000, 1)}))

    # 4) lexical FTS (fei_amd/memdir/fts_index.py): same docs, one bm25
    #    query instead of the scan
    import sqlite3
    db = sqlite3.connect(":memory:")
    db.execute("CREATE VIRTUAL TABLE f USING fts5(key UNINDEXED, content, "
               "tokenize='porter unicode61')")
    t0 = time.perf_counter()
    db.executemany("INSERT INTO f VALUES (?, ?)",
                   ((str(i), d) for i, d in enumerate(docs)))
    db.commit()
    build_s = time.perf_counter() - t0
    lat = []
    for _ in range(50):
        t0 = time.perf_counter()
        db.execute("SELECT key, bm25(f) r FROM f WHERE f MATCH ? "
                   "ORDER BY r LIMIT 10",
                   ('"kernel" "memory" "profile"',)).fetchall()
        lat.append((time.perf_counter() - t0) * 1000)
    lat.sort()
    print(json.dumps({"metric": "lexical FTS query (bm25 top-10, 3-term AND)",
                      "unit": "ms", "p50": round(lat[len(lat) // 2], 3),
                      "corpus": len(docs),
                      "build_s": round(build_s, 2),
                      "note": "the full-scan row is in-memory and flatters "
                              "the reference, which re-reads FILES per query"}))


if __name__ == "__main__":
    main()
