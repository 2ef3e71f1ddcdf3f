"""Interactive serving of a long-lived selector (reference: demo/app.py).

The reference ships a Gradio quiz wrapping the same L2/L3 API: CODA picks
the next point, a human labels it (or says "I don't know", which removes
the point without a label - demo/app.py:188-189), and live P(best) /
true-accuracy charts update after every answer. This module provides that
serving pattern as a FastAPI service (one full EIG pass per human answer -
the interactive-latency story) plus a small HTML
page; any HTTP client can drive the loop.

Endpoints:
    POST /start            reset the session (optional method/params)
    GET  /next             -> {"index": i, "prob": q}
    POST /answer           {"index": i, "label": c} -> updated state
    POST /skip             {"index": i}  ("I don't know")
    GET  /pbest            -> P(model is best) vector
    GET  /state            -> step, labeled count, best model, regret
"""
from __future__ import annotations

import random
import threading
from typing import Optional

import torch


class SelectorSession:
    """A long-lived selector driven step-by-step (thread-safe)."""

    def __init__(self, dataset, method: str = "coda", seed: int = 0,
                 oracle=None, **selector_kwargs):
        self.dataset = dataset
        self.method = method
        self.seed = seed
        self.oracle = oracle
        self.selector_kwargs = selector_kwargs
        self._lock = threading.Lock()
        self._pending: Optional[int] = None
        self.reset()

    def reset(self):
        from ..selectors import CODA
        from ..baselines import (IID, Uncertainty, ActiveTesting, VMA,
                                 ModelPicker)
        from ..options import LOSS_FNS
        with self._lock:
            random.seed(self.seed)
            torch.manual_seed(self.seed)
            loss = LOSS_FNS["acc"]
            m = self.method
            if m.startswith("coda"):
                self.selector = CODA(self.dataset, **self.selector_kwargs)
            elif m == "iid":
                self.selector = IID(self.dataset, loss)
            elif m == "uncertainty":
                self.selector = Uncertainty(self.dataset, loss)
            elif m == "activetesting":
                self.selector = ActiveTesting(self.dataset, loss)
            elif m == "vma":
                self.selector = VMA(self.dataset, loss)
            elif m == "model_picker":
                self.selector = ModelPicker(self.dataset)
            else:
                raise ValueError(m)
            self._pending = None
            self.n_answered = 0

    # -- step-wise protocol -------------------------------------------------
    def next_item(self):
        with self._lock:
            idx, prob = self.selector.get_next_item_to_label()
            self._pending = int(idx)
            return int(idx), float(prob)

    def answer(self, index: int, label: int):
        with self._lock:
            self.selector.add_label(int(index), int(label), 1.0)
            self._pending = None
            self.n_answered += 1
            return self._state_locked()

    def skip(self, index: int):
        """'I don't know': drop the point without labeling it."""
        with self._lock:
            if hasattr(self.selector, "skip"):
                self.selector.skip(int(index))
            else:
                unl = getattr(self.selector, "unlabeled_idxs", None)
                if unl is None:
                    unl = self.selector.d_u_idxs
                if int(index) in unl:
                    unl.remove(int(index))
            self._pending = None
            return self._state_locked()

    def pbest(self):
        with self._lock:
            sel = self.selector
            if hasattr(sel, "get_pbest"):
                return sel.get_pbest().detach().cpu().tolist()
            risk = sel.get_risk_estimates()
            inv = 1.0 / (risk + 1e-6)
            return (inv / inv.sum()).detach().cpu().tolist()

    def state(self):
        with self._lock:
            return self._state_locked()

    def _state_locked(self):
        best = int(self.selector.get_best_model_prediction())
        out = {"step": self.n_answered, "best_model": best,
               "n_labeled": self.n_answered,
               "pending": self._pending,
               "method": self.method}
        if self.oracle is not None:
            losses = self.oracle.true_losses(self.dataset.preds)
            out["true_best_model"] = int(losses.argmin())
            out["regret"] = float(losses[best] - losses.min())
            out["model_accuracies"] = (1 - losses).cpu().tolist()
        return out


_PAGE = """<!doctype html><html><head><title>coda_amd demo</title></head>
<body style="font-family:sans-serif;max-width:640px;margin:2em auto">
<h2>coda_amd - active model selection</h2>
<p>Point <b id="idx">?</b> - enter its true class label:</p>
<input id="label" type="number" min="0" style="width:6em">
<button onclick="answer()">Answer</button>
<button onclick="skip()">I don't know</button>
<pre id="state"></pre>
<script>
async function refresh(){
  const n = await (await fetch('/next')).json();
  document.getElementById('idx').textContent = n.index;
  const s = await (await fetch('/state')).json();
  document.getElementById('state').textContent = JSON.stringify(s, null, 2);
}
async function answer(){
  const idx = document.getElementById('idx').textContent;
  const label = document.getElementById('label').value;
  await fetch('/answer', {method:'POST',
    headers:{'Content-Type':'application/json'},
    body: JSON.stringify({index: +idx, label: +label})});
  refresh();
}
async function skip(){
  const idx = document.getElementById('idx').textContent;
  await fetch('/skip', {method:'POST',
    headers:{'Content-Type':'application/json'},
    body: JSON.stringify({index: +idx})});
  refresh();
}
refresh();
</script></body></html>"""


def create_app(dataset, method: str = "coda", oracle=None, seed: int = 0,
               images_dir: str = None, class_names=None,
               **selector_kwargs):
    """images_dir: optional folder whose sorted image files correspond to
    point indices (the order coda_amd.serve.build_predictions uses) - the
    reference demo's image-quiz flow (demo/app.py:137-172)."""
    from fastapi import Body, FastAPI, HTTPException
    from fastapi.responses import HTMLResponse

    session = SelectorSession(dataset, method=method, seed=seed,
                              oracle=oracle, **selector_kwargs)
    app = FastAPI(title="coda_amd serving")
    app.state.session = session

    image_paths = None
    if images_dir:
        from .build_predictions import list_images
        image_paths, _, detected = list_images(images_dir)
        if class_names is None:
            class_names = detected
    app.state.class_names = class_names

    @app.get("/", response_class=HTMLResponse)
    def index():
        return _PAGE

    @app.post("/start")
    def start(req: dict = Body(default={})):
        if req.get("method"):
            session.method = req["method"]
        if req.get("seed") is not None:
            session.seed = int(req["seed"])
        session.reset()
        return session.state()

    @app.get("/next")
    def next_item():
        idx, prob = session.next_item()
        return {"index": idx, "prob": prob}

    @app.post("/answer")
    def answer(req: dict = Body(...)):
        return session.answer(int(req["index"]), int(req["label"]))

    @app.post("/skip")
    def skip(req: dict = Body(...)):
        return session.skip(int(req["index"]))

    @app.get("/pbest")
    def pbest():
        return {"pbest": session.pbest()}

    @app.get("/pbest.png")
    def pbest_png():
        """Live P(best) bar chart (the reference demo's probability
        chart, demo/app.py:212-255)."""
        import io
        from fastapi.responses import Response
        from ..util import plot_bar
        img = plot_bar(session.pbest(), title="P(model is best)",
                       xlabel="model", ylabel="probability")
        buf = io.BytesIO()
        img.save(buf, format="PNG")
        return Response(content=buf.getvalue(), media_type="image/png")

    @app.get("/state")
    def state():
        st = session.state()
        if class_names:
            st["class_names"] = list(class_names)
        return st

    @app.get("/image/{index}")
    def image(index: int):
        """The point's image (sorted-order correspondence with the
        prediction tensor built by build_predictions)."""
        from fastapi.responses import FileResponse
        if image_paths is None:
            raise HTTPException(404, "no images_dir configured")
        if not 0 <= index < len(image_paths):
            raise HTTPException(404, "index out of range")
        return FileResponse(image_paths[index])

    return app


def main():
    import argparse
    import uvicorn
    from ..datasets import Dataset
    from ..oracle import Oracle
    from ..options import LOSS_FNS

    ap = argparse.ArgumentParser()
    ap.add_argument("--task", required=True)
    ap.add_argument("--data-dir", default="data")
    ap.add_argument("--method", default="coda")
    ap.add_argument("--images-dir", default=None,
                    help="serve point images from this folder")
    ap.add_argument("--device", default=None)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=7860)
    args = ap.parse_args()

    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    import os
    ds = Dataset(os.path.join(args.data_dir, args.task + ".pt"), device)
    oracle = Oracle(ds, LOSS_FNS["acc"]) if ds.labels is not None else None
    app = create_app(ds, method=args.method, oracle=oracle,
                     images_dir=args.images_dir)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
