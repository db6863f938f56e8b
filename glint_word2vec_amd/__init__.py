"""glint_word2vec_amd — MI355X-native skip-gram-with-negative-sampling
Word2Vec for very large vocabularies.

A from-scratch rebuild of the capabilities of MGabr/glint-word2vec
(Spark + Glint parameter servers) for a single 8xAMD MI355X node:
row-sharded embedding tables in HBM take the parameter-server role, RCCL
alltoallv over xGMI replaces Glint push/pull, and the SGNS inner loop is a
fused hand-written CDNA4 HIP kernel.  See SURVEY.md / BASELINE.md.
"""
from .config import Word2VecConfig
from .estimator import GlintWord2Vec, GlintWord2VecModel, LocalWord2VecModel
from .vocab import Vocabulary, build_vocab


def load_sharded(path, device="auto", dtype="auto"):
    """Convenience: GlintWord2VecModel.load_sharded (streaming,
    device-resident serving model)."""
    return GlintWord2VecModel.load_sharded(path, device=device, dtype=dtype)

__version__ = "0.1.0"
__all__ = [
    "Word2VecConfig", "GlintWord2Vec", "GlintWord2VecModel",
    "LocalWord2VecModel", "Vocabulary", "build_vocab", "load_sharded",
]
