"""g2vec_amd — an MI355X-native gene-embedding framework.

A from-scratch rebuild of the capabilities of mathcom/G2Vec
(reference: /root/reference/G2Vec.py) designed for AMD Instinct MI355X
(gfx950, CDNA4): the random-path generator is a CSR biased-random-walk
HIP kernel, the modified-CBOW trainer is a hand-written HIP kernel chain
(embedding gather -> wave reduce -> fused loss -> scatter-add backward ->
dense Adam), PCC graph construction offers an MFMA f32 GEMM path, and
data parallelism runs over RCCL/xGMI via torch.distributed.

Layer map (reference file:line -> module):
  CLI/config      G2Vec.py:505-518  -> g2vec_amd.cli / g2vec_amd.config
  Data I/O        G2Vec.py:436-503  -> g2vec_amd.io
  Preprocess      G2Vec.py:393-434  -> g2vec_amd.preprocess
  Graph (PCC)     G2Vec.py:354-391  -> g2vec_amd.graph
  Random walks    G2Vec.py:324-352  -> g2vec_amd.walks (+ ops HIP kernel)
  Path set        G2Vec.py:288-322  -> g2vec_amd.paths
  CBOW trainer    G2Vec.py:217-286  -> g2vec_amd.models.cbow
  L-groups        G2Vec.py:167-200  -> g2vec_amd.cluster
  Scoring         G2Vec.py:85-157   -> g2vec_amd.scoring
  Writers         G2Vec.py:120-215  -> g2vec_amd.io.writers
  (absent in ref) distributed       -> g2vec_amd.parallel
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
