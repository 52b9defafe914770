"""Model factory (ref: src/models/model_factory.{h,cpp}
createModelFromOptions)."""
from __future__ import annotations

from ..common.options import Options
from .transformer import TransformerModel


def create_model(options: Options):
    mtype = options.get("type", "transformer")
    if mtype in ("transformer", "transformer-encoder"):
        return TransformerModel(options)
    if mtype == "s2s":
        from .s2s import S2SModel
        return S2SModel(options)
    if mtype == "bert":
        from .bert import BertModel
        return BertModel(options)
    if mtype == "bert-classifier":
        from .bert import BertClassifier
        return BertClassifier(options)
    raise NotImplementedError(
        f"--type {mtype} is not implemented yet in marian_amd "
        "(transformer and s2s are)")
