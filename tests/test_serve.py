"""Serving layer: the interactive step-wise protocol over FastAPI."""
import os

import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from coda_amd.datasets import Dataset, make_synthetic_task  # noqa: E402
from coda_amd.oracle import Oracle  # noqa: E402
from coda_amd.options import LOSS_FNS  # noqa: E402
from coda_amd.serve import create_app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    preds, labels = make_synthetic_task(H=5, N=150, C=4, seed=3)
    ds = Dataset.from_tensors(preds, labels, "cpu")
    oracle = Oracle(ds, LOSS_FNS["acc"])
    app = create_app(ds, method="coda", oracle=oracle, chunk_size=64)
    return TestClient(app), oracle


def test_full_interactive_loop(client):
    c, oracle = client
    st = c.get("/state").json()
    assert st["step"] == 0
    for _ in range(3):
        nxt = c.get("/next").json()
        idx = nxt["index"]
        assert 0 <= idx < 150 and nxt["prob"] >= 0
        label = oracle(idx)
        st = c.post("/answer", json={"index": idx, "label": label}).json()
        assert "best_model" in st and "regret" in st
    assert st["step"] == 3
    pb = c.get("/pbest").json()["pbest"]
    assert len(pb) == 5
    assert abs(sum(pb) - 1.0) < 1e-3


def test_skip_removes_point(client):
    c, _ = client
    nxt = c.get("/next").json()
    before = c.get("/state").json()["step"]
    st = c.post("/skip", json={"index": nxt["index"]}).json()
    assert st["step"] == before  # no label consumed
    # the skipped point is never proposed again
    seen = {c.get("/next").json()["index"] for _ in range(5)}
    assert nxt["index"] not in seen


def test_restart_resets(client):
    c, _ = client
    st = c.post("/start", json={"method": "iid"}).json()
    assert st["step"] == 0 and st["method"] == "iid"
    nxt = c.get("/next").json()
    st = c.post("/answer", json={"index": nxt["index"], "label": 0}).json()
    assert st["step"] == 1
    # switch back to coda
    st = c.post("/start", json={"method": "coda"}).json()
    assert st["step"] == 0


def test_html_page(client):
    c, _ = client
    r = c.get("/")
    assert r.status_code == 200 and "coda_amd" in r.text


def test_synthetic_builder(tmp_path):
    import subprocess
    import sys
    import torch
    out = tmp_path / "demo.pt"
    r = subprocess.run(
        [sys.executable, "-m", "coda_amd.serve.build_predictions",
         "--synthetic", "4", "--n", "60", "--classes-n", "5",
         "--out", str(out)],
        capture_output=True, text=True, timeout=120,
        cwd="/root/repo")
    assert r.returncode == 0, r.stderr
    t = torch.load(str(out), weights_only=True)
    assert t.shape == (4, 60, 5)
    assert (tmp_path / "demo_labels.pt").exists()


def test_pbest_png(client):
    c, _ = client
    r = c.get("/pbest.png")
    assert r.status_code == 200
    assert r.headers["content-type"] == "image/png"
    assert r.content[:4] == b"\x89PNG"


def test_image_endpoint(tmp_path):
    from PIL import Image
    import numpy as np
    d = tmp_path / "imgs" / "cat"
    d.mkdir(parents=True)
    for i in range(3):
        Image.fromarray(
            (np.random.rand(8, 8, 3) * 255).astype("uint8")).save(
            d / f"im{i}.png")
    preds, labels = make_synthetic_task(H=3, N=3, C=2, seed=8)
    ds = Dataset.from_tensors(preds, labels, "cpu")
    app = create_app(ds, method="iid", images_dir=str(tmp_path / "imgs"))
    c = TestClient(app)
    r = c.get("/image/1")
    assert r.status_code == 200
    assert r.content[:4] == b"\x89PNG"
    assert c.get("/image/99").status_code == 404
    assert c.get("/state").json()["class_names"] == ["cat"]


@pytest.mark.gpu
def test_interactive_loop_gpu():
    """The serving protocol with the selector on cuda:0 (HIP kernels in
    the loop)."""
    import torch
    assert torch.cuda.is_available()
    preds, labels = make_synthetic_task(H=6, N=200, C=5, seed=9)
    ds = Dataset.from_tensors(preds, labels, "cuda:0")
    oracle = Oracle(ds, LOSS_FNS["acc"])
    app = create_app(ds, method="coda", oracle=oracle, chunk_size=64)
    c = TestClient(app)
    for _ in range(2):
        nxt = c.get("/next").json()
        st = c.post("/answer", json={"index": nxt["index"],
                                     "label": oracle(nxt["index"])}).json()
    assert st["step"] == 2
    pb = c.get("/pbest").json()["pbest"]
    assert len(pb) == 6 and abs(sum(pb) - 1.0) < 1e-3


def _make_tiny_clip(ckpt_dir: str):
    """Random-init CLIP checkpoint + minimal byte-level tokenizer, built
    entirely offline (the environment has no model weights): enough for
    the zero-shot pipeline to execute its real code path."""
    import json
    from transformers import (CLIPConfig, CLIPModel, CLIPTokenizer,
                              CLIPTextConfig, CLIPVisionConfig,
                              CLIPProcessor, CLIPImageProcessor)
    os.makedirs(ckpt_dir, exist_ok=True)
    vocab = {"<|startoftext|>": 0, "<|endoftext|>": 1}
    chars = "abcdefghijklmnopqrstuvwxyz "
    for i, ch in enumerate(chars):
        vocab[ch] = 2 + i
        vocab[ch + "</w>"] = 2 + len(chars) + i
    vp = os.path.join(ckpt_dir, "vocab.json")
    json.dump(vocab, open(vp, "w"))
    mp = os.path.join(ckpt_dir, "merges.txt")
    open(mp, "w").write("#version: 0.2\n")
    tok = CLIPTokenizer(vp, mp)
    cfg = CLIPConfig(
        text_config=CLIPTextConfig(
            vocab_size=len(vocab), hidden_size=32, intermediate_size=64,
            num_hidden_layers=2, num_attention_heads=2,
            max_position_embeddings=32, bos_token_id=0,
            eos_token_id=1).to_dict(),
        vision_config=CLIPVisionConfig(
            hidden_size=32, intermediate_size=64, num_hidden_layers=2,
            num_attention_heads=2, image_size=32,
            patch_size=8).to_dict(),
        projection_dim=16)
    CLIPModel(cfg).save_pretrained(ckpt_dir)
    CLIPProcessor(
        image_processor=CLIPImageProcessor(
            size={"shortest_edge": 32},
            crop_size={"height": 32, "width": 32}),
        tokenizer=tok).save_pretrained(ckpt_dir)


def test_hf_zero_shot_builder_end_to_end(tmp_path, monkeypatch):
    """The HF zero-shot model path of serve/build_predictions executes
    end-to-end offline (random-init tiny CLIP checkpoints; reference
    counterpart demo/hf_zeroshot.py:71-219): image-folder scan -> two
    zero-shot pipelines -> stacked (H, N, C) .pt + labels, loadable by
    the Dataset."""
    pytest.importorskip("transformers")
    from PIL import Image

    monkeypatch.setenv("HF_HUB_OFFLINE", "1")
    monkeypatch.setenv("TRANSFORMERS_OFFLINE", "1")
    imgdir = tmp_path / "imgs"
    for ci, cls in enumerate(["cat", "dog"]):
        (imgdir / cls).mkdir(parents=True)
        for j in range(3):
            Image.new("RGB", (32, 32),
                      (40 * ci + 20 * j, 80, 120)).save(
                imgdir / cls / f"{j}.png")
    ck1 = tmp_path / "m1"
    ck2 = tmp_path / "m2"
    _make_tiny_clip(str(ck1))
    _make_tiny_clip(str(ck2))

    out = tmp_path / "hf_task.pt"
    from coda_amd.serve import build_predictions as bp
    import sys
    argv = ["prog", "--images", str(imgdir),
            "--models", f"{ck1},{ck2}", "--out", str(out)]
    monkeypatch.setattr(sys, "argv", argv)
    bp.main()

    preds = torch.load(str(out), weights_only=False)
    assert preds.shape == (2, 6, 2)
    torch.testing.assert_close(preds.sum(-1),
                               torch.ones(2, 6), rtol=1e-4, atol=1e-4)
    labels = torch.load(str(out).replace(".pt", "_labels.pt"),
                        weights_only=False)
    assert labels.tolist() == [0, 0, 0, 1, 1, 1]

    from coda_amd.datasets import Dataset
    ds = Dataset(str(out), device="cpu")
    assert ds.preds.shape == (2, 6, 2) and ds.labels is not None
