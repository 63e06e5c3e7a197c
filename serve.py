"""Inference server over an exported model directory (C24 follow-through:
the reference exports a SavedModel for serving, train.py:246 — this is the
serving side of that capability).

    python serve.py --model_dir model --src_vocab src_vocab.txt \
        --tgt_vocab tgt_vocab.txt [--device cuda] [--port 8000]

Endpoints:
    GET  /health               -> {"status": "ok", "config": {...}}
    POST /translate            {"text": "...", "max_len": 10}
                               -> {"tokens": [...], "text": "..."}

Greedy KV-cached decode (models/transformer.py greedy_decode); the decode
runs under torch.no_grad on whatever device the model was loaded to.  On
CUDA, requests are padded into (src_len, max_len) shape buckets and served
by a hipGraph-captured decoder (models/transformer.py GraphedDecoder): one
graph replay per generated token instead of hundreds of eager launches —
the eager loop is launch-bound at B=1.  Padding the source is semantics-
preserving (pad keys are masked in every attention; pad rows never feed a
real query), so bucketed answers match the unpadded eager decode.
"""

from __future__ import annotations

import argparse

import torch


def build_app(model_dir: str, src_vocab: str, tgt_vocab: str,
              device: str = "cpu", dtype=None):
    from fastapi import Body, FastAPI

    from transformer_amd.runtime.export import load_exported
    from transformer_amd.data.tokenizer import SubwordTokenizer
    from transformer_amd.models.transformer import (GraphedDecoder,
                                                    greedy_decode)

    model, config = load_exported(model_dir, device=device, dtype=dtype)
    model.eval()
    src_tok = SubwordTokenizer.load_from_file(src_vocab)
    tgt_tok = SubwordTokenizer.load_from_file(tgt_vocab)

    use_graph = device.startswith("cuda") and torch.cuda.is_available()
    # Bucket cache is LRU-bounded and keys are clamped: each entry holds KV
    # caches + a captured graph, so request-controlled (src_len, max_len)
    # must not grow GPU memory without limit.
    from collections import OrderedDict
    MAX_DECODE_LEN = 256          # server ceiling on requested max_len
    MAX_SRC_LEN = 512             # requests longer than this are rejected
    MAX_GRAPH_BUCKETS = 8
    graphed = OrderedDict()  # (S_src_bucket, max_len_bucket) -> decoder
    # FastAPI runs sync endpoints in a threadpool; GraphedDecoder replays
    # mutate shared capture buffers, so decodes are serialized.
    import threading
    graph_lock = threading.Lock()

    def get_graphed(s_src: int, max_len: int):
        key = (max(16, (s_src + 15) // 16 * 16),
               max(16, (max_len + 15) // 16 * 16))
        if key in graphed:
            graphed.move_to_end(key)
        else:
            while len(graphed) >= MAX_GRAPH_BUCKETS:
                graphed.popitem(last=False)
            graphed[key] = GraphedDecoder(
                model, B=1, S_src=key[0], max_len=key[1],
                start_id=tgt_tok.vocab_size,
                device=next(model.parameters()).device)
        return graphed[key], key[0]

    app = FastAPI(title="transformer_amd", version="1.0")

    @app.get("/health")
    def health():
        return {"status": "ok", "config": config}

    @app.post("/translate")
    def translate(payload: dict = Body(...)):
        text = str(payload.get("text", ""))
        max_len = min(max(int(payload.get("max_len", 10)), 1),
                      MAX_DECODE_LEN)
        src_start = src_tok.vocab_size
        tokens = [src_start] + src_tok.encode(text) + [src_start + 1]
        if len(tokens) > MAX_SRC_LEN:
            return {"error": f"source too long (> {MAX_SRC_LEN} tokens)"}
        enc_in = torch.tensor([tokens], dtype=torch.int64,
                              device=next(model.parameters()).device)
        tgt_start = tgt_tok.vocab_size
        with torch.no_grad():
            if use_graph:
                with graph_lock:
                    dec, s_bucket = get_graphed(len(tokens), max_len)
                    padded = torch.zeros(1, s_bucket, dtype=torch.int64,
                                         device=enc_in.device)
                    padded[0, :len(tokens)] = enc_in[0]
                    out = dec(padded, tgt_start + 1, max_len=max_len)
            else:
                out = greedy_decode(model, enc_in, tgt_start, tgt_start + 1,
                                    max_len=max_len)
        ids = out.squeeze(0).tolist()
        body = [t for t in ids if 0 < t < tgt_tok.vocab_size]
        return {"tokens": ids, "text": tgt_tok.decode(body)}

    return app


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model_dir", type=str, default="model")
    p.add_argument("--src_vocab", type=str, default="src_vocab.txt")
    p.add_argument("--tgt_vocab", type=str, default="tgt_vocab.txt")
    p.add_argument("--device", type=str,
                   default="cuda" if torch.cuda.is_available() else "cpu")
    p.add_argument("--host", type=str, default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args(argv)

    import uvicorn
    app = build_app(args.model_dir, args.src_vocab, args.tgt_vocab,
                    args.device,
                    torch.bfloat16 if args.device.startswith("cuda") else None)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
