"""HTTP serving for job results — query a finished MapReduce job without
materializing it.

The reference's consumption story ends at finalfn / result files
(server.lua:348-385: read every result.P file to do anything).  On the
GPU tier results are HBM-resident and indexed, so point queries are
device-side lookups (InvertedIndexResult.lookup, WordCountResult.topk)
— this module exposes them over HTTP for serving deployments:

    from mapreduce_amd.serve import make_app
    app = make_app(wordcount=res, index=idx)        # results stay on GPU
    uvicorn.run(app, host="0.0.0.0", port=8000)

Endpoints:
    GET /healthz                     liveness + which results are mounted
    GET /count?word=the              count of one word (binary search)
    GET /topk?k=10                   k most frequent words
    GET /postings?word=the           inverted-index postings [(doc, tf)]
"""

from __future__ import annotations

from typing import Optional


def make_app(wordcount=None, index=None):
    """Build a FastAPI app over a WordCountResult and/or
    InvertedIndexResult (either tier; tensors stay where they are)."""
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="mapreduce_amd results")
    state = {"wordcount": wordcount, "index": index}

    def _wc():
        if state["wordcount"] is None:
            raise HTTPException(404, "no wordcount result mounted")
        return state["wordcount"]

    def _ix():
        if state["index"] is None:
            raise HTTPException(404, "no index result mounted")
        return state["index"]

    @app.get("/healthz")
    def healthz():
        return {"ok": True,
                "wordcount": state["wordcount"] is not None,
                "index": state["index"] is not None}

    @app.get("/count")
    def count(word: str):
        return {"word": word, "count": _wc().count_of(word)}

    @app.get("/topk")
    def topk(k: int = 10):
        res = _wc()
        return {"topk": [{"word": w.decode("utf-8", "replace"), "count": c}
                         for w, c in res.topk(k)]}

    @app.get("/postings")
    def postings(word: str):
        res = _ix()
        return {"word": word,
                "postings": [{"doc": d, "tf": t}
                             for d, t in res.lookup(word)]}

    return app


def serve(wordcount=None, index=None, host: str = "127.0.0.1",
          port: int = 8000, app=None) -> None:
    """Run the result server (blocking)."""
    import uvicorn

    uvicorn.run(app or make_app(wordcount, index), host=host, port=port)


def build_results_from_files(paths, device=None):
    """Run wordcount + inverted index over files and return the two
    result objects (the CLI's one-stop build; GPU if available)."""
    import torch

    from .gpu.input import load_corpus
    from .gpu.inverted_index import InvertedIndexJob
    from .gpu.wordcount import WordCountJob

    dev = device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    corpus = load_corpus(list(paths), dev)
    splits = corpus.splits()
    wc = WordCountJob(dev, vocab_estimate=1 << 17).run(corpus.text, splits)
    ix = InvertedIndexJob(dev).run(corpus.text, splits)
    return wc, ix


def main(argv: Optional[list] = None) -> int:
    """CLI: index files and serve queries.

        python -m mapreduce_amd.serve doc1.txt doc2.txt --port 8000
    """
    import argparse

    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("files", nargs="+")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    p.add_argument("--device", default=None)
    args = p.parse_args(argv)
    wc, ix = build_results_from_files(args.files, args.device)
    print(f"# serving {wc.nwords or '?'} words, "
          f"{wc.keys.numel()} unique, {len(args.files)} docs "
          f"on {args.host}:{args.port}")
    serve(wc, ix, host=args.host, port=args.port)
    return 0


if __name__ == "__main__":
    import sys

    sys.exit(main())
