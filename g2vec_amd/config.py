"""Run configuration.

Mirrors the reference CLI surface (G2Vec.py:505-518) and adds the
MI355X-framework knobs (device, dtype, seed, DP) that default to
reference-equivalent behaviour.
"""
from __future__ import annotations

import dataclasses
from typing import Optional


@dataclasses.dataclass
class G2VecConfig:
    # --- reference-compatible surface (G2Vec.py:505-518) ---
    expression_file: str = ""
    clinical_file: str = ""
    network_file: str = ""
    result_name: str = "result"
    len_path: int = 80          # -p / --lenPath
    num_repetition: int = 10    # -r / --numRepetition
    hidden: int = 128           # -s / --sizeHiddenlayer
    epochs: int = 500           # -e / --epoch (honoured here; ref ignores it, G2Vec.py:262)
    lr: float = 0.005           # -l / --learningRate
    num_biomarker: int = 50     # -n / --numBiomarker

    # --- framework knobs (absent in reference) ---
    seed: Optional[int] = 0         # None -> nondeterministic like the reference
    dtype: str = "fp32"             # W_ih storage dtype of the GENERAL kernel
                                    # chain's gather ({"fp32","bf16","fp16"};
                                    # fp32 master weights/grads either way).
                                    # SCOPE: trainer_path="general" only — the
                                    # default fast path's collapsed scalar
                                    # algebra always computes in fp32, i.e. AT
                                    # OR ABOVE the requested precision.
                                    # validate() rejects a reduced dtype
                                    # combined with the fast path so the flag
                                    # can never silently do nothing.
    device: str = "auto"            # "auto" | "cpu" | "cuda"
    pcc_threshold: float = 0.5      # |PCC| cutoff (G2Vec.py:385-390)
    pcc_mode: str = "auto"          # "edge" (per-edge dot) | "gemm" (MFMA corr GEMM) | "auto"
    kmeans_backend: str = "auto"    # "sklearn" (reference parity) | "torch" (scales) | "auto"
    compat_lgroup_bug: bool = False  # reproduce the shipped G2Vec.py:186-194 behaviour
    early_stop: bool = True
    earlystop_every: int = 1        # early-stop accuracy readback granularity:
                                    # 1 = per-epoch (reference semantics and
                                    # schedule); k>1 = read accuracies every k
                                    # epochs and deterministically REPLAY to
                                    # the dip on stop — same trajectory/stop/
                                    # weights, 1/k the collectives + D2H
                                    # (opt-in; see cbow.run_epochs_kgranular)
    batch_size: int = 0             # 0 = full batch (reference semantics, G2Vec.py:264)
    trainer_path: str = "fast"      # "fast": collapsed rank-1 path (linear-net algebra)
                                    # "general": full gather/scatter kernel chain (K1-K8)
    activation: str = "none"        # hidden activation: "none" (reference
                                    # semantics, G2Vec.py:238-240) | "relu"
                                    # (opt-in non-linear successor on the
                                    # GENERAL chain; the fast path's linear
                                    # collapse does not apply, so validate()
                                    # requires --trainer-path general)
    save_paths: str = ""            # cache generated path set (de-facto checkpoint)
    save_model: str = ""            # save trained W_ih/W_ho + metadata (.pt)
    train_ckpt: str = ""            # mid-TRAINING checkpoint file: weights +
                                    # Adam moments + early-stop trackers,
                                    # written every train_ckpt_every epochs;
                                    # --resume-train continues the EXACT
                                    # trajectory (epoch bodies deterministic).
                                    # Runs the synchronous epoch loop
                                    # (checkpoints are epoch-aligned).
    train_ckpt_every: int = 5
    resume_train: str = ""          # resume training from a train_ckpt file
    load_model: str = ""            # resume: skip step 4, load W_ih from .pt
    load_paths: str = ""
    log_jsonl: str = ""             # structured metrics sink
    gene_relabel: str = "auto"      # trainer-internal gene-id relabeling for
                                    # gather locality: "on" | "off" | "auto"
                                    # (auto = on at n_genes >= 100k, where the
                                    # random s-gathers hit the L2 random-line
                                    # wall). First-touch order over the path
                                    # set makes co-path genes contiguous, so
                                    # consecutive paths read L1-resident
                                    # s/W slices. Pure layout change: weights
                                    # are un-permuted before results leave the
                                    # trainer; outputs are unchanged up to
                                    # fp32 reduction order.
    use_hipgraph: bool = True       # record the full-batch epoch into a hipGraph

    def validate(self) -> None:
        if self.hidden not in (64, 128, 256, 512, 1024):
            raise ValueError(
                f"hidden={self.hidden}: MI355X kernels require a power-of-two "
                f"multiple of the 64-lane wavefront in {{64,128,256,512,1024}}")
        if self.dtype not in ("fp32", "bf16", "fp16"):
            raise ValueError(f"dtype must be fp32|bf16|fp16, got {self.dtype}")
        if self.dtype != "fp32" and self.trainer_path != "general":
            raise ValueError(
                f"dtype={self.dtype} applies to the general kernel chain's "
                f"W_ih gather storage only; the fast path computes in fp32 "
                f"regardless. Use --trainer-path general with a reduced "
                f"dtype, or drop --dtype.")
        if self.train_ckpt_every < 1:
            raise ValueError("train_ckpt_every must be >= 1")
        if self.earlystop_every < 1:
            raise ValueError("earlystop_every must be >= 1")
        if self.epochs < 1:
            raise ValueError("epochs must be >= 1 (the reference always runs "
                             "at least one epoch, G2Vec.py:262)")
        if self.len_path < 1 or self.len_path > 512:
            raise ValueError("len_path must be in [1, 512] (LDS visited-list budget)")
        # note: total path nnz (sum of path lengths) is limited to < 2^31
        # by the int32 kernel index space — enforced at PathSet build time
        # (paths.py guards integrate_pathsets/subset)
        if self.pcc_mode not in ("auto", "edge", "gemm"):
            raise ValueError(f"bad pcc_mode {self.pcc_mode}")
        if self.trainer_path not in ("fast", "general"):
            raise ValueError(f"bad trainer_path {self.trainer_path}")
        if self.activation not in ("none", "relu"):
            raise ValueError(f"bad activation {self.activation}")
        if self.activation != "none" and self.trainer_path != "general":
            raise ValueError(
                "activation != none requires --trainer-path general: the "
                "fast path exploits the reference net's LINEARITY "
                "(o = sum s_g), which a hidden activation breaks")
        if self.kmeans_backend not in ("auto", "sklearn", "torch"):
            raise ValueError(f"bad kmeans_backend {self.kmeans_backend}")
        if self.gene_relabel not in ("auto", "on", "off"):
            raise ValueError(f"bad gene_relabel {self.gene_relabel}")


def resolve_device(device: str) -> str:
    if device != "auto":
        return device
    import torch
    return "cuda" if torch.cuda.is_available() else "cpu"
