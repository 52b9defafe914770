"""marian-conv: model format conversion (ref: src/command/marian_conv.cpp).

Supported: .npz <-> .bin (the mmap-able flat binary format for fast
inference loads), dtype casting (fp32/fp16/bf16 stored as fp32+tag),
shortlist text -> binary.
"""
import argparse
import json
import struct
import sys

import numpy as np

from ..common import io_npz

BIN_MAGIC = b"MAMDBIN1"

_DT = {"float32": 0, "float16": 1, "bfloat16": 2, "int8": 3}
_DT_REV = {v: k for k, v in _DT.items()}
_NP = {"float32": np.float32, "float16": np.float16,
       "bfloat16": np.float32,  # stored widened (numpy has no bf16)
       "int8": np.int8}


def save_bin(path, items, config):
    """Flat binary: header JSON + aligned raw tensors (mmap-friendly),
    standing in for the reference's binary.cpp format."""
    header = {"config": config or {}, "tensors": []}
    blobs = []
    off = 0
    for name, arr in items.items():
        arr = np.ascontiguousarray(arr)
        dt = str(arr.dtype)
        if dt not in _DT:
            arr = arr.astype(np.float32)
            dt = "float32"
        pad = (-off) % 64
        off += pad
        header["tensors"].append({"name": name, "shape": list(arr.shape),
                                  "dtype": dt, "offset": off,
                                  "bytes": arr.nbytes})
        blobs.append((pad, arr))
        off += arr.nbytes
    hjson = json.dumps(header).encode("utf-8")
    with open(path, "wb") as f:
        f.write(BIN_MAGIC)
        f.write(struct.pack("<Q", len(hjson)))
        f.write(hjson)
        base = f.tell()
        pad0 = (-base) % 64
        f.write(b"\0" * pad0)
        for pad, arr in blobs:
            f.write(b"\0" * pad)
            f.write(arr.tobytes())


def load_bin(path):
    with open(path, "rb") as f:
        magic = f.read(8)
        assert magic == BIN_MAGIC, "not a marian_amd .bin file"
        (hlen,) = struct.unpack("<Q", f.read(8))
        header = json.loads(f.read(hlen).decode("utf-8"))
        base = f.tell()
        base += (-base) % 64
        data = np.fromfile(f, dtype=np.uint8, offset=0)
    items = {}
    for t in header["tensors"]:
        start = base - (16 + hlen) + t["offset"]
        raw = data[start:start + t["bytes"]]
        arr = raw.view(_NP[t["dtype"]]).reshape(t["shape"])
        items[t["name"]] = arr.copy()
    return items, header.get("config")


def main(argv=None):
    p = argparse.ArgumentParser(prog="marian-conv")
    p.add_argument("--from", dest="src", required=True)
    p.add_argument("--to", dest="dst", required=True)
    p.add_argument("--gemm-type", default="float32",
                   choices=list(_DT.keys()))
    p.add_argument("--shortlist", nargs="+", default=[],
                   help="binarize a lexical shortlist: "
                        "--shortlist DIC SRC_VOCAB TRG_VOCAB "
                        "(--from/--to are the dic in/out paths)")
    p.add_argument("--export-as", default="",
                   help="onnx: export encoder+decoder graphs "
                        "(<to>.encoder.onnx / <to>.decoder.onnx, "
                        "ref: marian-conv --export-as onnx)")
    args = p.parse_args(argv)

    if args.export_as:
        if args.export_as not in ("onnx", "onnx-encode"):
            raise SystemExit(f"unknown --export-as {args.export_as}")
        from ..common.options import Options
        from ..models.model_factory import create_model
        items, cfg = io_npz.load_items(args.src)
        if not cfg:
            raise SystemExit("model has no embedded config "
                             "(special:model.yml) - cannot rebuild")
        opts = Options(dict(cfg))
        model = create_model(opts)
        model.load_marian_items(items, strict=False)
        from ..translator.onnx_export import export_onnx
        vocab = (opts.list("dim-vocabs") or [32000])[-1]
        prefix = args.dst[:-5] if args.dst.endswith(".onnx") else args.dst
        files = export_onnx(model, prefix, int(vocab))
        print("Exported " + ", ".join(files))
        return 0

    if args.shortlist:
        # text .dic -> id-resolved .blex.npz (ref: marian-conv --shortlist)
        from ..data.vocab import Vocab
        dic = args.shortlist[0] if args.shortlist else args.src
        sv = Vocab.load(args.shortlist[1])
        tv = Vocab.load(args.shortlist[2])
        src_ids, trg_ids, probs = [], [], []
        with open(dic, encoding="utf-8") as f:
            for line in f:
                parts = line.rstrip("\n").split("\t")
                if len(parts) < 2:
                    parts = line.split()
                if len(parts) < 2:
                    continue
                sid = sv.token2id.get(parts[0])
                tid = tv.token2id.get(parts[1])
                if sid is None or tid is None:
                    continue
                src_ids.append(sid)
                trg_ids.append(tid)
                probs.append(float(parts[2]) if len(parts) > 2 else 1.0)
        np.savez(args.dst, src=np.asarray(src_ids, dtype=np.int64),
                 trg=np.asarray(trg_ids, dtype=np.int64),
                 prob=np.asarray(probs, dtype=np.float32))
        print(f"Binarized shortlist {dic} -> {args.dst} "
              f"({len(src_ids)} pairs)")
        return

    if args.src.endswith(".npz"):
        items, cfg = io_npz.load_items(args.src)
    elif args.src.endswith(".bin"):
        items, cfg = load_bin(args.src)
    else:
        raise SystemExit(f"unknown input format: {args.src}")

    if args.gemm_type != "float32":
        items = {k: (v.astype(np.float16) if args.gemm_type == "float16"
                     and v.dtype == np.float32 else v)
                 for k, v in items.items()}

    if args.dst.endswith(".npz"):
        io_npz.save_items(args.dst, {k: np.asarray(v) for k, v in items.items()},
                          cfg)
    elif args.dst.endswith(".bin"):
        save_bin(args.dst, items, cfg)
    else:
        raise SystemExit(f"unknown output format: {args.dst}")
    print(f"Converted {args.src} -> {args.dst}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
