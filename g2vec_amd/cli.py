"""CLI — reference-compatible surface (G2Vec.py:505-518): the published
command lines (`python -m g2vec_amd EXPR CLIN NET NAME -p 80 -r 10 ...`)
work unchanged; framework flags are additive and default to
reference-equivalent behaviour."""
from __future__ import annotations

import argparse

from .config import G2VecConfig
from .parallel.dist import init_dist
from .pipeline import run


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(
        prog="g2vec_amd",
        description="MI355X-native network-based identification of prognostic "
                    "gene signatures (capabilities of mathcom/G2Vec).")
    p.add_argument("EXPRESSION_FILE", type=str,
                   help="Tab-delimited gene expression profiles.")
    p.add_argument("CLINICAL_FILE", type=str,
                   help="Tab-delimited clinical data; LABEL 0=good, 1=poor.")
    p.add_argument("NETWORK_FILE", type=str,
                   help="Tab-delimited gene interaction network.")
    p.add_argument("RESULT_NAME", type=str,
                   help="Prefix for *_biomarkers.txt, *_lgroups.txt, *_vectors.txt")
    p.add_argument("-p", "--lenPath", type=int, default=80)
    p.add_argument("-r", "--numRepetition", type=int, default=10)
    p.add_argument("-s", "--sizeHiddenlayer", type=int, default=128)
    p.add_argument("-e", "--epoch", type=int, default=500,
                   help="honoured here (the reference parses but ignores it)")
    p.add_argument("-l", "--learningRate", type=float, default=0.005)
    p.add_argument("-n", "--numBiomarker", type=int, default=50)
    # framework flags (absent in reference)
    p.add_argument("--seed", type=int, default=0,
                   help="-1 = unseeded (reference-like nondeterminism)")
    p.add_argument("--dtype", choices=["fp32", "bf16", "fp16"], default="fp32",
               help="W_ih gather storage dtype, general kernel chain only "
                    "(the fast path always computes fp32); reduced dtypes "
                    "require --trainer-path general")
    p.add_argument("--device", choices=["auto", "cpu", "cuda"], default="auto")
    p.add_argument("--pcc-mode", choices=["auto", "edge", "gemm"], default="auto")
    p.add_argument("--pcc-threshold", type=float, default=0.5,
                   help="|PCC| cutoff for graph edges (reference: 0.5)")
    p.add_argument("--kmeans", choices=["auto", "sklearn", "torch"],
                   default="auto", help="L-group clustering backend")
    p.add_argument("--activation", choices=["none", "relu"], default="none",
                   help="hidden activation (general chain only; the fast "
                        "path exploits the reference net's linearity)")
    p.add_argument("--trainer-path", choices=["fast", "general"], default="fast")
    p.add_argument("--batch-size", type=int, default=0,
                   help="0 = full batch (reference semantics)")
    p.add_argument("--compat-lgroup-bug", action="store_true",
                   help="reproduce the shipped L-group disambiguation bug "
                        "(G2Vec.py:186-194, SURVEY 2.9)")
    p.add_argument("--no-early-stop", action="store_true")
    p.add_argument("--train-ckpt", default="",
                   help="mid-training checkpoint file (weights + Adam "
                        "moments + early-stop trackers), written every "
                        "--train-ckpt-every epochs; resume with "
                        "--resume-train for an exact continuation")
    p.add_argument("--train-ckpt-every", type=int, default=5)
    p.add_argument("--resume-train", default="")
    p.add_argument("--earlystop-every", type=int, default=1,
                   help="read early-stop accuracies every K epochs and "
                        "deterministically replay to the dip on stop "
                        "(same trajectory/stop/weights; 1/K the "
                        "collectives and readbacks; fast path)")
    p.add_argument("--gene-relabel", choices=["auto", "on", "off"],
                   default="auto",
                   help="trainer-internal gather-locality gene relabeling "
                        "(auto = on at >= 100k genes; pure layout change)")
    p.add_argument("--no-hipgraph", action="store_true",
                   help="disable hipGraph capture of the training epoch")
    p.add_argument("--save-paths", type=str, default="")
    p.add_argument("--load-model", type=str, default="",
                   help="skip training: load W_ih from a --save-model .pt "
                        "(gene count and hidden size must match)")
    p.add_argument("--save-model", type=str, default="",
                   help="save trained weights + metadata as a .pt checkpoint")
    p.add_argument("--load-paths", type=str, default="")
    p.add_argument("--log-jsonl", type=str, default="")
    return p


def args_to_config(a: argparse.Namespace) -> G2VecConfig:
    return G2VecConfig(
        expression_file=a.EXPRESSION_FILE, clinical_file=a.CLINICAL_FILE,
        network_file=a.NETWORK_FILE, result_name=a.RESULT_NAME,
        len_path=a.lenPath, num_repetition=a.numRepetition,
        hidden=a.sizeHiddenlayer, epochs=a.epoch, lr=a.learningRate,
        num_biomarker=a.numBiomarker,
        seed=(None if a.seed < 0 else a.seed), dtype=a.dtype, device=a.device,
        pcc_mode=a.pcc_mode, pcc_threshold=a.pcc_threshold,
        kmeans_backend=a.kmeans, trainer_path=a.trainer_path,
        activation=a.activation, gene_relabel=a.gene_relabel,
        batch_size=a.batch_size, compat_lgroup_bug=a.compat_lgroup_bug,
        early_stop=not a.no_early_stop, earlystop_every=a.earlystop_every,
        train_ckpt=a.train_ckpt, train_ckpt_every=a.train_ckpt_every,
        resume_train=a.resume_train,
        save_paths=a.save_paths,
        load_paths=a.load_paths, save_model=a.save_model,
        load_model=a.load_model,
        log_jsonl=a.log_jsonl,
        use_hipgraph=not a.no_hipgraph)


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    cfg = args_to_config(args)
    ctx = init_dist(cfg.device)
    try:
        run(cfg, ctx)
    finally:
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
