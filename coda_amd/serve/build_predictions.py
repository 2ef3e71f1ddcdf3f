"""Build (H, N, C) prediction tensors from a pool of zero-shot classifiers.

Equivalent of the reference's demo/hf_zeroshot.py (CLIP/SigLIP/BioCLIP
zero-shot inference over demo images -> per-model score files): runs a
pool of HuggingFace zero-shot image-classification checkpoints over a
local image folder and stacks their post-softmax scores into the
framework's on-disk (H, N, C) .pt format (+ optional _labels.pt from the
folder structure).

Requires locally cached model weights (this environment has no network);
pass --models with paths or hub ids resolvable from the local HF cache.
`--synthetic H` builds a synthetic pool instead (no weights needed).

Usage:
    python -m coda_amd.serve.build_predictions --images dir/ \
        --classes "cat,dog,bird" --models path1,path2 --out data/mytask.pt
    python -m coda_amd.serve.build_predictions --synthetic 8 \
        --n 500 --classes-n 10 --out data/demo.pt
"""
from __future__ import annotations

import argparse
import os

import torch


def list_images(root: str):
    """(paths, labels, class_names): labels from subfolder names when the
    folder is structured imagenet-style, else None."""
    exts = (".jpg", ".jpeg", ".png", ".bmp", ".webp")
    subdirs = sorted(d for d in os.listdir(root)
                     if os.path.isdir(os.path.join(root, d)))
    paths, labels = [], []
    if subdirs:
        for ci, d in enumerate(subdirs):
            for f in sorted(os.listdir(os.path.join(root, d))):
                if f.lower().endswith(exts):
                    paths.append(os.path.join(root, d, f))
                    labels.append(ci)
        return paths, labels, subdirs
    for f in sorted(os.listdir(root)):
        if f.lower().endswith(exts):
            paths.append(os.path.join(root, f))
    return paths, None, None


def run_zero_shot(model_id: str, image_paths, class_names, device,
                  batch_size: int = 16) -> torch.Tensor:
    """(N, C) post-softmax scores from one zero-shot checkpoint."""
    from PIL import Image
    from transformers import pipeline

    pipe = pipeline("zero-shot-image-classification", model=model_id,
                    device=0 if device.type == "cuda" else -1)
    out = torch.zeros(len(image_paths), len(class_names))
    for i0 in range(0, len(image_paths), batch_size):
        batch = [Image.open(p).convert("RGB")
                 for p in image_paths[i0:i0 + batch_size]]
        results = pipe(batch, candidate_labels=list(class_names))
        if isinstance(results[0], dict):
            results = [results]
        for j, res in enumerate(results):
            for entry in res:
                c = class_names.index(entry["label"])
                out[i0 + j, c] = entry["score"]
    return out / out.sum(-1, keepdim=True).clamp_min(1e-12)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--images", default=None, help="image folder")
    ap.add_argument("--classes", default=None,
                    help="comma-separated class names (default: subfolders)")
    ap.add_argument("--models", default=None,
                    help="comma-separated HF checkpoints (local paths or "
                         "cached hub ids)")
    ap.add_argument("--synthetic", type=int, default=0,
                    help="build a synthetic H-model pool instead")
    ap.add_argument("--n", type=int, default=500)
    ap.add_argument("--classes-n", type=int, default=10)
    ap.add_argument("--out", required=True)
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)

    if args.synthetic:
        from ..datasets import make_synthetic_task
        preds, labels = make_synthetic_task(H=args.synthetic, N=args.n,
                                            C=args.classes_n)
        torch.save(preds, args.out)
        torch.save(labels, args.out.replace(".pt", "_labels.pt"))
        print(f"wrote synthetic pool {tuple(preds.shape)} -> {args.out}")
        return

    assert args.images and args.models, "--images and --models required"
    device = torch.device(args.device or
                          ("cuda" if torch.cuda.is_available() else "cpu"))
    paths, labels, subdirs = list_images(args.images)
    class_names = args.classes.split(",") if args.classes else subdirs
    assert class_names, "give --classes or use a class-subfolder layout"
    print(f"{len(paths)} images, {len(class_names)} classes")

    per_model = []
    for model_id in args.models.split(","):
        print("running", model_id)
        per_model.append(run_zero_shot(model_id, paths, class_names, device))
    preds = torch.stack(per_model)          # (H, N, C)
    torch.save(preds, args.out)
    if labels is not None:
        torch.save(torch.tensor(labels), args.out.replace(".pt",
                                                          "_labels.pt"))
    print(f"wrote {tuple(preds.shape)} -> {args.out}")


if __name__ == "__main__":
    main()
