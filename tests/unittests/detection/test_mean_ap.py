"""MeanAveragePrecision: reference doctest oracles + segm/RLE tests."""
import pytest
import torch

from metrics_amd.detection import MeanAveragePrecision




def test_map_bbox_reference_doc_example():
    """Reference detection/mean_ap.py:247 doctest values."""
    m = MeanAveragePrecision(iou_type="bbox")
    m.update(
        [dict(boxes=torch.tensor([[258.0, 41.0, 606.0, 285.0]]), scores=torch.tensor([0.536]), labels=torch.tensor([0]))],
        [dict(boxes=torch.tensor([[214.0, 41.0, 562.0, 285.0]]), labels=torch.tensor([0]))],
    )
    r = m.compute()
    assert abs(float(r["map"]) - 0.6) < 1e-6
    assert float(r["map_50"]) == 1.0 and float(r["map_75"]) == 1.0
    assert abs(float(r["map_large"]) - 0.6) < 1e-6
    assert float(r["map_small"]) == -1.0 and float(r["map_medium"]) == -1.0
    assert abs(float(r["mar_1"]) - 0.6) < 1e-6


def test_map_segm_reference_doc_example():
    """Reference detection/mean_ap.py:287 doctest values (mask IoU = 3/5)."""
    mask_pred = torch.zeros(5, 5, dtype=torch.bool)
    mask_pred[1:3, 2:4] = True
    mask_tgt = torch.zeros(5, 5, dtype=torch.bool)
    mask_tgt[1:4, 2] = True
    mask_tgt[2, 3] = True
    m = MeanAveragePrecision(iou_type="segm")
    m.update(
        [dict(masks=mask_pred[None], scores=torch.tensor([0.536]), labels=torch.tensor([0]))],
        [dict(masks=mask_tgt[None], labels=torch.tensor([0]))],
    )
    r = m.compute()
    assert abs(float(r["map"]) - 0.2) < 1e-6
    assert float(r["map_50"]) == 1.0 and float(r["map_75"]) == 0.0
    assert abs(float(r["mar_100"]) - 0.2) < 1e-6


def test_map_segm_perfect_and_miss():
    big = torch.zeros(20, 20, dtype=torch.bool)
    big[0:10, 0:10] = True
    shifted = torch.zeros(20, 20, dtype=torch.bool)
    shifted[0:10, 5:15] = True  # IoU = 50/150 = 1/3 < 0.5
    m = MeanAveragePrecision(iou_type="segm")
    m.update(
        [dict(masks=big[None], scores=torch.tensor([0.9]), labels=torch.tensor([1]))],
        [dict(masks=big[None], labels=torch.tensor([1]))],
    )
    assert abs(float(m.compute()["map"]) - 1.0) < 1e-6
    m2 = MeanAveragePrecision(iou_type="segm")
    m2.update(
        [dict(masks=shifted[None], scores=torch.tensor([0.9]), labels=torch.tensor([1]))],
        [dict(masks=big[None], labels=torch.tensor([1]))],
    )
    assert float(m2.compute()["map"]) == 0.0


def test_map_bbox_and_segm_tuple():
    mask_pred = torch.zeros(5, 5, dtype=torch.bool)
    mask_pred[1:3, 2:4] = True
    mask_tgt = torch.zeros(5, 5, dtype=torch.bool)
    mask_tgt[1:4, 2] = True
    mask_tgt[2, 3] = True
    box = torch.tensor([[2.0, 1.0, 4.0, 3.0]])
    m = MeanAveragePrecision(iou_type=("bbox", "segm"))
    m.update(
        [dict(boxes=box, masks=mask_pred[None], scores=torch.tensor([0.536]), labels=torch.tensor([0]))],
        [dict(boxes=box, masks=mask_tgt[None], labels=torch.tensor([0]))],
    )
    r = m.compute()
    assert abs(float(r["bbox_map"]) - 1.0) < 1e-6  # identical boxes
    assert abs(float(r["segm_map"]) - 0.2) < 1e-6
    assert "classes" in r and "bbox_mar_100" in r and "segm_mar_100" in r


def test_map_segm_multi_instance_and_crowd():
    a = torch.zeros(16, 16, dtype=torch.bool); a[0:4, 0:4] = True
    b = torch.zeros(16, 16, dtype=torch.bool); b[8:12, 8:12] = True
    crowd_gt = torch.zeros(16, 16, dtype=torch.bool); crowd_gt[0:16, 0:2] = True
    det_in_crowd = torch.zeros(16, 16, dtype=torch.bool); det_in_crowd[2:6, 0:2] = True
    m = MeanAveragePrecision(iou_type="segm")
    m.update(
        [dict(masks=torch.stack([a, b, det_in_crowd]), scores=torch.tensor([0.9, 0.8, 0.7]),
              labels=torch.tensor([0, 0, 0]))],
        [dict(masks=torch.stack([a, b, crowd_gt]), labels=torch.tensor([0, 0, 0]),
              iscrowd=torch.tensor([0, 0, 1]))],
    )
    r = m.compute()
    # both real gts matched perfectly; crowd det fully inside crowd gt is ignored
    assert abs(float(r["map"]) - 1.0) < 1e-6


def test_map_segm_rle_roundtrip():
    from metrics_amd.detection.mean_ap import _encode_masks_rle, _decode_masks_rle

    torch.manual_seed(11)
    masks = torch.rand(4, 13, 17) > 0.5
    pack = _encode_masks_rle(masks)
    dec = _decode_masks_rle(pack.numpy())
    ref = masks.transpose(1, 2).reshape(4, -1).numpy()  # column-major
    assert (dec == ref).all()
    empty = _encode_masks_rle(torch.zeros(0, 5, 5, dtype=torch.bool))
    assert _decode_masks_rle(empty.numpy()).shape == (0, 25)


def test_tm_to_coco_round_trip(tmp_path):
    """tm_to_coco -> coco_to_tm reproduces boxes/masks (compressed-RLE strings)."""
    import torch
    from torch import tensor

    from metrics_amd.detection import MeanAveragePrecision

    name = str(tmp_path / "rt")
    m = MeanAveragePrecision(iou_type="bbox")
    m.update(
        [dict(boxes=tensor([[258.0, 41.0, 606.0, 285.0]]), scores=tensor([0.536]), labels=tensor([0]))],
        [dict(boxes=tensor([[214.0, 41.0, 562.0, 285.0]]), labels=tensor([0]))],
    )
    m.tm_to_coco(name)
    preds, target = MeanAveragePrecision.coco_to_tm(f"{name}_preds.json", f"{name}_target.json", iou_type="bbox")
    assert torch.allclose(preds[0]["boxes"], tensor([[258.0, 41.0, 606.0, 285.0]]))
    assert torch.allclose(target[0]["boxes"], tensor([[214.0, 41.0, 562.0, 285.0]]))
    assert preds[0]["scores"].item() == pytest.approx(0.536)

    masks = torch.zeros(2, 10, 12, dtype=torch.uint8)
    masks[0, 2:6, 3:9] = 1
    masks[1, 1:3, 1:3] = 1
    ms = MeanAveragePrecision(iou_type="segm")
    ms.update(
        [dict(masks=masks, scores=tensor([0.7, 0.3]), labels=tensor([1, 2]))],
        [dict(masks=masks, labels=tensor([1, 2]))],
    )
    ms.tm_to_coco(str(tmp_path / "seg"))
    ps, ts = MeanAveragePrecision.coco_to_tm(
        str(tmp_path / "seg_preds.json"), str(tmp_path / "seg_target.json"), iou_type="segm"
    )
    assert (ps[0]["masks"] == masks).all()
    assert (ts[0]["masks"] == masks).all()


def test_coco_rle_string_codec():
    """Compressed-RLE string encode/decode round-trips arbitrary run lists."""
    import numpy as np

    from metrics_amd.detection.mean_ap import _coco_rle_str_decode, _coco_rle_str_encode

    rng = np.random.default_rng(0)
    for _ in range(20):
        counts = rng.integers(0, 10_000, size=rng.integers(1, 40)).tolist()
        assert _coco_rle_str_decode(_coco_rle_str_encode(counts)) == counts


def test_segm_rle_dict_inputs_match_tensor_masks():
    """update() accepts pycocotools-style RLE dicts (compressed string or
    plain counts list) and scores identically to dense tensor masks."""
    import numpy as np

    from metrics_amd.detection.mean_ap import _coco_rle_str_encode

    torch.manual_seed(77)

    def to_rle_dicts(masks, compress):
        out = []
        for m in masks.numpy():
            flat = np.asfortranarray(m).T.reshape(-1)  # column-major
            change = np.flatnonzero(flat[1:] != flat[:-1]) + 1
            idx = np.concatenate([[0], change, [flat.size]])
            counts = (idx[1:] - idx[:-1]).tolist()
            if flat[0]:
                counts = [0] + counts
            c = _coco_rle_str_encode(counts) if compress else counts
            out.append({"size": [m.shape[0], m.shape[1]], "counts": c})
        return out

    H = W = 24
    preds_t, tgts_t, preds_r, tgts_r = [], [], [], []
    for i in range(3):
        nd, ng = 4, 3
        dmasks = (torch.rand(nd, H, W) > 0.6).to(torch.uint8)
        gmasks = (torch.rand(ng, H, W) > 0.6).to(torch.uint8)
        scores = torch.rand(nd)
        dl = torch.randint(0, 2, (nd,))
        gl = torch.randint(0, 2, (ng,))
        preds_t.append({"masks": dmasks.bool(), "scores": scores, "labels": dl})
        tgts_t.append({"masks": gmasks.bool(), "labels": gl})
        preds_r.append({"masks": to_rle_dicts(dmasks, compress=i % 2 == 0), "scores": scores, "labels": dl})
        tgts_r.append({"masks": to_rle_dicts(gmasks, compress=i % 2 == 1), "labels": gl})

    mt = MeanAveragePrecision(iou_type="segm")
    mt.update(preds_t, tgts_t)
    rt = mt.compute()
    mr = MeanAveragePrecision(iou_type="segm")
    mr.update(preds_r, tgts_r)
    rr = mr.compute()
    for k in ("map", "map_50", "mar_100"):
        assert torch.allclose(rt[k], rr[k], atol=1e-6), k
