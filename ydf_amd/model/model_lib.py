"""Model persistence (reference model/model_library.h:44-58 SaveModel /
LoadModel registry)."""
from __future__ import annotations

import json
import os

import numpy as np

from ydf_amd.dataset.dataspec import DataSpecification, Task
from ydf_amd.model.forest import FlatForest
from ydf_amd.model.generic_model import GenericModel


def load_model(path: str) -> GenericModel:
    from ydf_amd.model.specialized import MODEL_CLASSES

    if not os.path.exists(os.path.join(path, "header.json")):
        if os.path.exists(os.path.join(path, "header.pb")):
            # reference-format directory (the default save() layout):
            # forest/dataspec from the wire files, framework-only state
            # from extra.json (absent when the dir came from the
            # reference itself)
            from ydf_amd.model.import_ydf import load_ydf_model

            model = load_ydf_model(path)
            ej = os.path.join(path, "extra.json")
            if os.path.exists(ej):
                with open(ej) as f:
                    extra = json.load(f)
                model.activation = extra.get("activation",
                                             model.activation)
                meta = dict(extra.get("metadata") or {})
                meta.setdefault("loss", (model.metadata or {}).get("loss"))
                model.metadata = meta
                model.training_logs = extra.get("training_logs")
                model._load_extra(extra)
            return model
        raise FileNotFoundError(
            f"{path}: neither header.json (npz container) nor header.pb "
            f"(reference model directory) found")
    with open(os.path.join(path, "header.json")) as f:
        header = json.load(f)
    if header.get("model_type") in ("MLP", "TABULAR_TRANSFORMER"):
        from ydf_amd.deep import DeepModel

        return DeepModel.load(path)
    with open(os.path.join(path, "dataspec.json")) as f:
        dataspec = DataSpecification.from_json(json.load(f))
    z = np.load(os.path.join(path, "forest.npz"))
    forest = FlatForest(feat=z["feat"], thr=z["thr"], left=z["left"],
                        roots=z["roots"],
                        cat_idx=z["cat_idx"] if "cat_idx" in z else None,
                        masks=z["masks"] if "masks" in z else None,
                        cover=z["cover"] if "cover" in z else None,
                        obl_ranges=z["obl_ranges"] if "obl_ranges" in z
                        else None,
                        obl_attr=z["obl_attr"] if "obl_attr" in z else None,
                        obl_w=z["obl_w"] if "obl_w" in z else None,
                        na_right=z["na_right"] if "na_right" in z
                        else None,
                        set_idx=z["set_idx"] if "set_idx" in z else None,
                        set_offs=z["set_offs"] if "set_offs" in z
                        else None,
                        set_items=z["set_items"] if "set_items" in z
                        else None)
    cls = MODEL_CLASSES.get(header["model_type"], GenericModel)
    model = cls(
        forest=forest,
        dataspec=dataspec,
        task=Task[header["task"]],
        label_classes=header.get("label_classes"),
        init_predictions=header.get("init_predictions", [0.0]),
        num_trees_per_iter=header.get("num_trees_per_iter", 1),
        activation=header.get("activation", "identity"),
        metadata=header.get("metadata"),
    )
    model.training_logs = header.get("training_logs")
    model._load_extra(header)
    return model


def serialize_model(model: GenericModel) -> bytes:
    """In-memory serialization (mirrors ydf model.serialize())."""
    import io
    import zipfile

    buf = io.BytesIO()
    with zipfile.ZipFile(buf, "w") as zf:
        zf.writestr("header.json", json.dumps(model._header()))
        zf.writestr("dataspec.json", json.dumps(model.dataspec.to_json()))
        fbuf = io.BytesIO()
        np.savez(fbuf, feat=model.forest.feat, thr=model.forest.thr,
                 left=model.forest.left, roots=model.forest.roots,
                 cat_idx=model.forest.cat_idx, masks=model.forest.masks,
                 cover=model.forest.cover,
                 obl_ranges=model.forest.obl_ranges,
                 obl_attr=model.forest.obl_attr, obl_w=model.forest.obl_w,
                 na_right=model.forest.na_right,
                 set_idx=model.forest.set_idx,
                 set_offs=model.forest.set_offs,
                 set_items=model.forest.set_items)
        zf.writestr("forest.npz", fbuf.getvalue())
    return buf.getvalue()


def deserialize_model(data: bytes) -> GenericModel:
    import io
    import tempfile
    import zipfile

    with tempfile.TemporaryDirectory() as td:
        with zipfile.ZipFile(io.BytesIO(data)) as zf:
            zf.extractall(td)
        return load_model(td)
