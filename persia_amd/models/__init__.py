"""Built-in dense model families for recommender training.

All models take ``(non_id_type_tensors, embedding_tensors)`` like the
reference example models (examples/src/adult-income/model.py:25-27) so they
plug directly into ``ctx.forward``.
"""
from persia_amd.models.ctr import CTRModel
from persia_amd.models.dlrm import DLRM
from persia_amd.models.dcn import DCNv2

__all__ = ["CTRModel", "DLRM", "DCNv2"]
