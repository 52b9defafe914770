"""Model factory (ref: src/models/model_factory.{h,cpp}
createModelFromOptions)."""
from __future__ import annotations

from ..common.options import Options
from .transformer import TransformerModel


def create_model(options: Options):
    mtype = options.get("type", "transformer")
    if mtype in ("transformer", "transformer-encoder"):
        return TransformerModel(options)
    if mtype in ("s2s", "amun", "nematus"):
        from .s2s import S2SModel
        m = S2SModel(options)
        # amun/nematus = s2s computation + GroundHog-era weight names
        # (ref: src/models/amun.h, nematus.h)
        m.legacy_names = mtype in ("amun", "nematus")
        return m
    if mtype == "char-s2s":
        from .char_s2s import CharS2SModel
        return CharS2SModel(options)
    if mtype in ("lm", "lm-transformer"):
        from .lm import TransformerLM
        return TransformerLM(options)
    if mtype == "bert":
        from .bert import BertModel
        return BertModel(options)
    if mtype == "bert-classifier":
        from .bert import BertClassifier
        return BertClassifier(options)
    raise NotImplementedError(
        f"--type {mtype} is not implemented yet in marian_amd "
        "(transformer and s2s are)")
