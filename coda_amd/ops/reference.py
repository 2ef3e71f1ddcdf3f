"""Reference (eager PyTorch) implementations of every compute op.

These define the exact numerical semantics of the framework's compute core:
fp32 throughout, the clamp ladder (1e-6 / 1e-12 / 1e-30, log-space clamp
+-80), log2 entropies, and the Beta grid on [1e-6, 1-1e-6] with P=256 points.
Behavioral parity targets in the reference implementation (justinkay/coda):
  - consensus                -> coda/util.py:14
  - confusion_prior          -> coda/coda.py:28-43
  - init_dirichlets          -> coda/coda.py:46-63 (+ multiplier at :196)
  - dirichlet_to_beta        -> coda/coda.py:14-25
  - pi_hat_update            -> coda/coda.py:226-233
  - pbest_from_beta          -> coda/coda.py:77-119 (K6-K9)
  - hypothetical_betas       -> coda/coda.py:150-168 (K10)
  - eig_assemble             -> coda/coda.py:253-278 (K11)
  - disagreement_mask        -> coda/coda.py:215-224 (K12; simplified:
                                "some model disagrees with the majority" ==
                                "not all models agree", which is equivalent)
  - accuracy_losses          -> coda/oracle.py:9-21 + coda/options.py:3-14
  - entropy_acquisition      -> coda/baselines/uncertainty.py:6-11
  - vma_pairwise             -> coda/baselines/vma.py:31-41, computed via the
                                sorted-order identity (O(H log H) per point
                                instead of the reference's O(H^2) broadcast)
  - lure_weights             -> coda/baselines/activetesting.py:52-67

The CPU path runs these directly; the GPU path runs hand-written CDNA4 HIP
kernels validated against these (see coda_amd/ops/__init__.py dispatch).
"""
from __future__ import annotations

import torch

# Grid constants for the Beta P(best) integral (coda/coda.py:80,86)
PBEST_NUM_POINTS = 256
GRID_LO = 1e-6
GRID_HI = 1.0 - 1e-6
EPS_PROB = 1e-30
LOG_CLAMP = 80.0


# ---------------------------------------------------------------------------
# Prior construction
# ---------------------------------------------------------------------------

def consensus(preds: torch.Tensor) -> torch.Tensor:
    """Mean over the model axis: (H, N, C) -> (N, C), fp32."""
    return init_model_stats(preds)[1] / preds.shape[0]


def init_model_stats(preds: torch.Tensor, chunk_h: int = 0):
    """One chunked fp32-upcast pass over the model axis at selector init:
    cached argmax classes (H, N) + the consensus sum (N, C) fp32.
    Storage may be fp32 / bf16 / fp8 (coda_amd.datasets.STORAGE_DTYPES);
    compute is always fp32. chunk_h defaults to <= 1 GiB of fp32 chunk
    (a fixed 16 was a 64 GB transient at the 1M-point config)."""
    H, N, C = preds.shape
    if chunk_h <= 0:
        chunk_h = max(1, min(16, (1 << 28) // max(1, N * C)))
    classes = torch.empty(H, N, dtype=torch.long, device=preds.device)
    ens = torch.zeros(N, C, dtype=torch.float32, device=preds.device)
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        ch = preds[h0:h1].float()
        classes[h0:h1] = ch.argmax(-1)
        ens += ch.sum(0)
    return classes, ens


def confusion_prior(pseudo_labels: torch.Tensor, preds: torch.Tensor) -> torch.Tensor:
    """Soft confusion matrices from pseudo-labels: (N,), (H,N,C) -> (H,C,C).

    conf[h, c, j] = sum_{n: pseudo_labels[n]==c} preds[h, n, j], row-normalized
    with clamp_min(1e-6). Implemented as a segment-sum (index_add) rather than
    the one-hot einsum, which never materializes the (N, C) one-hot.
    """
    H, N, C = preds.shape
    conf = torch.zeros(H, C, C, dtype=torch.float32, device=preds.device)
    # Flatten (H, C) target rows: row index = h*C + label[n]; chunk over H
    # with fp32 upcast so bf16/fp8 storage accumulates exactly.
    chunk_h = max(1, min(H, (1 << 28) // max(1, N * C)))
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        idx = (pseudo_labels.unsqueeze(0) +
               torch.arange(h0, h1, device=preds.device).unsqueeze(1) * C)
        conf.view(H * C, C).index_add_(0, idx.reshape(-1),
                                       preds[h0:h1].float().reshape(-1, C))
    return conf / conf.sum(-1, keepdim=True).clamp_min(1e-6)


def init_dirichlets(soft_confusion: torch.Tensor,
                    prior_strength: float,
                    disable_diag_prior: bool = False,
                    multiplier: float = 1.0) -> torch.Tensor:
    """(H,C,C) prior: base (diag=1, off-diag=1/(C-1); or uniform 2/C) plus
    prior_strength * soft_confusion, all scaled by multiplier."""
    H, C, _ = soft_confusion.shape
    dev, dt = soft_confusion.device, soft_confusion.dtype
    if disable_diag_prior:
        base = torch.full((C, C), 2.0 / C, dtype=dt, device=dev)
    else:
        base = torch.full((C, C), 1.0 / (C - 1), dtype=dt, device=dev)
        base.fill_diagonal_(1.0)
    return multiplier * (base.unsqueeze(0) + prior_strength * soft_confusion)


def dirichlet_to_beta(dirichlets: torch.Tensor):
    """(..., C, C) -> diagonal Beta params (alpha, beta), each (..., C)."""
    C = dirichlets.shape[-1]
    diag_idx = torch.arange(C, device=dirichlets.device)
    alpha = dirichlets[..., diag_idx, diag_idx]
    beta = dirichlets.sum(dim=-1) - alpha
    return alpha, beta


# ---------------------------------------------------------------------------
# pi_hat (confusion-adjusted class marginals)
# ---------------------------------------------------------------------------

def pi_hat_partial(dirichlets: torch.Tensor, preds: torch.Tensor,
                   chunk_h: int = 32) -> torch.Tensor:
    """Unnormalized per-item class scores summed over the LOCAL model axis:
    sum_h preds[h] @ dirichlets[h]^T -> (N, C) fp32.

    adjusted[h,n,c] = sum_s dirichlets[h,c,s] * preds[h,n,s]; the H sum is the
    RCCL all-reduce site when the model axis is sharded.

    `preds` may be bf16 (the MI355X fast path: bf16 MFMA at ~16x the f32
    rate; per-GEMM results are accumulated into fp32 across models, so only
    the per-model contraction rounds at bf16). The fp32 path matches the
    reference bit-for-bit semantics (coda/coda.py:227).
    """
    H = preds.shape[0]
    out = None
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        part = torch.bmm(preds[h0:h1].float(),
                         dirichlets[h0:h1].float().transpose(1, 2)).sum(0)
        out = part if out is None else out + part
    return out


def pi_hat_pack_chunked(preds: torch.Tensor, chunk_h: int = 16):
    """pi_hat_pack for bf16/fp8 storage: fills the (N, H*C) bf16 operand
    model-by-model without materializing a permuted fp32 copy."""
    H, N, C = preds.shape
    packed = torch.empty(N, H * C, dtype=torch.bfloat16,
                         device=preds.device)
    view = packed.view(N, H, C)
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        view[:, h0:h1] = preds[h0:h1].to(torch.bfloat16).permute(1, 0, 2)
    return packed


def pi_hat_pack(preds: torch.Tensor) -> torch.Tensor:
    """Pack (H, N, C) predictions as a (N, H*C) bf16 GEMM operand.

    pi_hat's contraction adjusted[n,c] = sum_h sum_s preds[h,n,s]*D[h,c,s]
    is a SINGLE GEMM of shape (N) x (C) x (K=H*C): A[n, h*C+s] =
    preds[h,n,s], B[h*C+s, c] = D[h,c,s]. Packing A once at selector init
    turns the per-step pi_hat update into one bf16 MFMA GEMM (f32
    accumulate over the whole K, which folds the model-axis sum into the
    GEMM - no (H,N,C) intermediate, ~16x the f32 matrix rate).

    Note: torch.bmm over the model batch is both slower (f32) and broken
    (hipBLASLt bf16 batched GEMM faults at batch=32, N=50k, C=1000 on
    ROCm 7.0) - hence this packed single-GEMM formulation.
    """
    H, N, C = preds.shape
    return preds.permute(1, 0, 2).reshape(N, H * C).contiguous().to(
        torch.bfloat16)


def pi_hat_partial_packed(dirichlets: torch.Tensor,
                          packed: torch.Tensor) -> torch.Tensor:
    """(H,C,C) Dirichlets + packed (N, H*C) bf16 preds -> (N, C) fp32."""
    H, C, _ = dirichlets.shape
    B = dirichlets.transpose(1, 2).reshape(H * C, C).to(packed.dtype)
    return (packed @ B).float()


def pi_hat_partial_streamed(dirichlets: torch.Tensor, preds: torch.Tensor,
                            chunk_n: int = 32768) -> torch.Tensor:
    """pi_hat_partial_packed without ever holding the full (N, H*C)
    operand: pack chunk_n points, GEMM, reuse the buffer.

    Peak extra memory is one (chunk_n, H*C) bf16 buffer (8 GB at
    H=128 x C=1000) instead of N x H x C x 2 bytes - 256 GB at the
    1M-point config, which cannot coexist with the prediction pool in
    288 GB HBM. Same math and the same bf16 MFMA GEMM rate (each chunk
    is still a huge (chunk_n) x (C) x (K=H*C) GEMM).
    """
    H, N, C = preds.shape
    B = dirichlets.transpose(1, 2).reshape(H * C, C).to(torch.bfloat16)
    out = torch.empty(N, C, dtype=torch.float32, device=preds.device)
    m = min(chunk_n, N)
    buf = torch.empty(m, H * C, dtype=torch.bfloat16, device=preds.device)
    for n0 in range(0, N, chunk_n):
        n1 = min(n0 + chunk_n, N)
        v = buf[: n1 - n0].view(n1 - n0, H, C)
        for h0 in range(0, H, 16):
            h1 = min(h0 + 16, H)
            v[:, h0:h1] = preds[h0:h1, n0:n1].to(
                torch.bfloat16).permute(1, 0, 2)
        out[n0:n1] = (buf[: n1 - n0] @ B).float()
    return out


def pi_hat_delta(preds: torch.Tensor, point_classes: torch.Tensor,
                 chunk_h: int = 32) -> torch.Tensor:
    """Incremental pi_hat update term for one labeled point.

    add_label moves ONLY row `true_class` of each model's Dirichlet
    (coda/coda.py:316-317: D[h, y, s] += lr*[s == argmax_h(idx)]), so the
    posterior marginal's change is rank-1:
        adjusted[n, y] += lr * sum_h preds[h, n, cls_h]
    with cls_h = model h's argmax class at the labeled point. This op
    returns sum_h preds[h, :, cls_h] -> (N,) fp32: an O(H*N) gather-sum
    replacing the O(H*N*C^2) full contraction per step (exact in exact
    arithmetic; fp32 += drift is ~1e-7/step).

    preds: (H, N, C) any storage dtype; point_classes: (H,) int64.
    """
    H, N, C = preds.shape
    out = torch.zeros(N, dtype=torch.float32, device=preds.device)
    idx = point_classes.view(H, 1, 1).expand(H, N, 1)
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        out += preds[h0:h1].float().gather(2, idx[h0:h1]).squeeze(-1).sum(0)
    return out


def pi_hat_normalize(adjusted_sum: torch.Tensor):
    """(N, C) unnormalized -> (pi_hat_xi (N,C), pi_hat (C,))."""
    pi_xi = adjusted_sum / adjusted_sum.sum(dim=-1, keepdim=True).clamp_min(1e-12)
    pi = pi_xi.sum(0)
    pi = pi / pi.sum()
    return pi_xi, pi


# ---------------------------------------------------------------------------
# The hot op: P(model is best) from diagonal Betas
# ---------------------------------------------------------------------------

def _beta_grid(device, dtype=torch.float32, num_points: int = PBEST_NUM_POINTS):
    return torch.linspace(GRID_LO, GRID_HI, num_points, device=device, dtype=dtype)


def beta_grid_pdf_cdf(alpha: torch.Tensor, beta: torch.Tensor,
                      num_points: int = PBEST_NUM_POINTS):
    """Beta pdf on the grid plus its trapezoid cumulative integral.

    alpha, beta: (R, H). Returns pdf, cdf: (R, H, P) and the grid x: (P,).
    """
    x = _beta_grid(alpha.device, alpha.dtype, num_points)
    lx = torch.log(x)
    l1mx = torch.log1p(-x)
    log_norm = (torch.lgamma(alpha) + torch.lgamma(beta)
                - torch.lgamma(alpha + beta))          # (R, H)
    logpdf = ((alpha.unsqueeze(-1) - 1.0) * lx
              + (beta.unsqueeze(-1) - 1.0) * l1mx
              - log_norm.unsqueeze(-1))                # (R, H, P)
    pdf = logpdf.exp()
    # Trapezoid cumulative integral with cdf[..., 0] = 0 (matches the
    # reference's sequential loop at coda/coda.py:98-101).
    cdf = torch.cat([torch.zeros_like(pdf[..., :1]),
                     torch.cumulative_trapezoid(pdf, x, dim=-1)], dim=-1)
    return pdf, cdf, x


def pbest_from_beta(alpha: torch.Tensor, beta: torch.Tensor,
                    num_points: int = PBEST_NUM_POINTS,
                    return_unnormalized: bool = False):
    """P(h has the largest Beta-distributed accuracy) per row.

    alpha, beta: (R, H) -> (R, H). For each row r:
      p_h = integral pdf_h(x) * prod_{h' != h} cdf_{h'}(x) dx, normalized over H.
    """
    pdf, cdf, x = beta_grid_pdf_cdf(alpha, beta, num_points)
    log_cdf = torch.log(cdf.clamp_min(EPS_PROB))
    prod_excl = torch.exp(
        (log_cdf.sum(dim=1, keepdim=True) - log_cdf).clamp(-LOG_CLAMP, LOG_CLAMP))
    integrand = pdf * prod_excl
    prob = torch.trapz(integrand, x, dim=-1)           # (R, H)
    if return_unnormalized:
        return prob
    return prob / prob.sum(-1, keepdim=True).clamp_min(EPS_PROB)


def pbest_from_beta_hchunked(alpha: torch.Tensor, beta: torch.Tensor,
                             num_points: int = PBEST_NUM_POINTS,
                             chunk_h: int = 512) -> torch.Tensor:
    """pbest_from_beta for very wide model axes: two passes chunked over
    H, so only (R, chunk_h, P) materializes (the plain eager form would
    need the full (R, H, P)). Used when H exceeds the HIP kernel's LDS
    budget (~2048 models/row)."""
    R, H = alpha.shape
    x = _beta_grid(alpha.device, alpha.dtype, num_points)
    slog = torch.zeros(R, num_points, dtype=alpha.dtype,
                       device=alpha.device)
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        _, cdf, _ = beta_grid_pdf_cdf(alpha[:, h0:h1], beta[:, h0:h1],
                                      num_points)
        slog += torch.log(cdf.clamp_min(EPS_PROB)).sum(dim=1)
    prob = torch.empty(R, H, dtype=alpha.dtype, device=alpha.device)
    for h0 in range(0, H, chunk_h):
        h1 = min(h0 + chunk_h, H)
        pdf, cdf, _ = beta_grid_pdf_cdf(alpha[:, h0:h1], beta[:, h0:h1],
                                        num_points)
        log_cdf = torch.log(cdf.clamp_min(EPS_PROB))
        pe = torch.exp((slog.unsqueeze(1) - log_cdf)
                       .clamp(-LOG_CLAMP, LOG_CLAMP))
        prob[:, h0:h1] = torch.trapz(pdf * pe, x, dim=-1)
    return prob / prob.sum(-1, keepdim=True).clamp_min(EPS_PROB)


def hypothetical_betas(alpha_cc: torch.Tensor, beta_cc: torch.Tensor,
                       pred_classes: torch.Tensor, update_weight: float = 1.0):
    """Hypothetical Beta updates for a candidate chunk.

    alpha_cc, beta_cc: (H, C) current diagonal Betas.
    pred_classes: (B, H) each model's argmax class on each candidate.
    Returns alpha, beta: (B, C, H) - for candidate b and hypothesized class c,
    model h's Beta after alpha += w if argmax==c else beta += w.
    """
    B, H = pred_classes.shape
    C = alpha_cc.shape[-1]
    eq = (pred_classes.unsqueeze(1) ==
          torch.arange(C, device=alpha_cc.device).view(1, C, 1))  # (B, C, H)
    a = alpha_cc.t().unsqueeze(0) + update_weight * eq.to(alpha_cc.dtype)
    b = beta_cc.t().unsqueeze(0) + update_weight * (~eq).to(beta_cc.dtype)
    return a, b


def mixture_entropy(pbest_rows: torch.Tensor, pi_hat: torch.Tensor):
    """mixture0 (H,) = sum_c pi_hat[c] * pbest_rows[c, h]; H0 = log2 entropy."""
    mixture0 = (pi_hat.unsqueeze(-1) * pbest_rows).sum(0)   # (H,)
    m = mixture0.clamp_min(1e-12)
    H0 = -(m * m.log2()).sum()
    return mixture0, H0


def eig_assemble(pbest_hyp: torch.Tensor, pbest_before: torch.Tensor,
                 pi_hat: torch.Tensor, pi_hat_xi_chunk: torch.Tensor,
                 mixture0: torch.Tensor, H_before: torch.Tensor) -> torch.Tensor:
    """Expected information gain per candidate.

    pbest_hyp: (B, C, H); pbest_before: (C, H); pi_hat: (C,);
    pi_hat_xi_chunk: (B, C); mixture0: (H,); H_before: scalar. -> (B,)
    """
    deltas = pi_hat.view(1, -1, 1) * (pbest_hyp - pbest_before.unsqueeze(0))
    mix_new = mixture0.view(1, 1, -1) + deltas            # (B, C, H)
    m = mix_new.clamp_min(1e-12)
    H_after = -(m * m.log2()).sum(-1)                     # (B, C)
    return H_before - (pi_hat_xi_chunk * H_after).sum(-1)


# ---------------------------------------------------------------------------
# Acquisition / evaluation helpers
# ---------------------------------------------------------------------------

def pred_classes(preds: torch.Tensor) -> torch.Tensor:
    """Argmax over classes: (H, N, C) -> (H, N) int64. Cached by callers.
    Up-casts low-precision storage dtypes that lack an argmax kernel."""
    try:
        return preds.argmax(dim=-1)
    except RuntimeError:
        return init_model_stats(preds)[0]


def disagreement_mask(classes: torch.Tensor) -> torch.Tensor:
    """(H, N) argmax classes -> (N,) bool: True where not all models agree."""
    return (classes != classes[0:1]).any(dim=0)


def accuracy_losses(classes: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """Mean (1 - accuracy) per model: (H, N) classes, (N,) labels -> (H,)."""
    return 1.0 - (classes == labels.unsqueeze(0)).float().mean(dim=1)


def entropy_acquisition(consensus_preds: torch.Tensor) -> torch.Tensor:
    """Ensemble-entropy per point: (N, C) -> (N,), natural log, eps=1e-8."""
    p = consensus_preds
    return -(p * torch.log(p + 1e-8)).sum(dim=-1)


def vma_pairwise(losses: torch.Tensor) -> torch.Tensor:
    """sum_{h'>h} |loss_h - loss_h'| per point via the sorted identity.

    losses: (H, M) -> (M,). For sorted x_(0) <= ... <= x_(H-1):
    sum_{i<j} (x_(j) - x_(i)) = sum_k (2k - H + 1) * x_(k).
    """
    H = losses.shape[0]
    s, _ = losses.sort(dim=0)
    w = (2.0 * torch.arange(H, device=losses.device, dtype=losses.dtype)
         - (H - 1))
    return (w.unsqueeze(-1) * s).sum(dim=0)


def lure_weights(qs: torch.Tensor, N: int) -> torch.Tensor:
    """LURE weights v_m = 1 + (N-M)/(N-m) * (1/((N-m+1) q_m) - 1), m 1-indexed.

    qs: (M,) sampling probabilities -> (M,).
    """
    M = qs.shape[0]
    m = torch.arange(1, M + 1, device=qs.device, dtype=qs.dtype)
    return 1.0 + ((N - M) / (N - m)) * (1.0 / ((N - m + 1.0) * qs) - 1.0)
