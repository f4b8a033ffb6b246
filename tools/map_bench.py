"""Config-4 benchmark: MeanAveragePrecision on COCO-shape synthetic boxes
(5000 images, up to 100 dets / 20 gts per image, 80 classes).

Run on GPU: ours (HIP IoU + OpenMP matcher + numpy accumulate).
Run with --reference (CPU, needs /root/reference): the reference's pure-torch
legacy COCOeval via the offline oracle loader.
"""
import argparse
import sys
import time

sys.path.insert(0, ".")
import torch


def make_data(n_img, seed=0):
    g = torch.Generator().manual_seed(seed)
    preds, target = [], []
    for _ in range(n_img):
        n_det = int(torch.randint(50, 101, (1,), generator=g))
        n_gt = int(torch.randint(5, 21, (1,), generator=g))
        pb = torch.rand(n_det, 4, generator=g) * 200
        pb[:, 2:] += pb[:, :2] + 5
        gb = torch.rand(n_gt, 4, generator=g) * 200
        gb[:, 2:] += gb[:, :2] + 5
        preds.append({
            "boxes": pb, "scores": torch.rand(n_det, generator=g),
            "labels": torch.randint(0, 80, (n_det,), generator=g),
        })
        target.append({"boxes": gb, "labels": torch.randint(0, 80, (n_gt,), generator=g)})
    return preds, target


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--images", type=int, default=5000)
    ap.add_argument("--batch", type=int, default=100)
    ap.add_argument("--reference", action="store_true")
    args = ap.parse_args()
    preds, target = make_data(args.images)

    if args.reference:
        sys.path.insert(0, "tests/unittests/detection")
        from _ref_oracle import load_legacy_map

        cls = load_legacy_map()
        if cls is None:
            print("reference not available here")
            return
        m = cls(iou_type="bbox")
    else:
        import metrics_amd as ma

        m = ma.detection.MeanAveragePrecision(iou_type="bbox")
        if torch.cuda.is_available():
            preds = [{k: v.cuda() for k, v in p.items()} for p in preds]
            target = [{k: v.cuda() for k, v in t.items()} for t in target]

    t0 = time.perf_counter()
    for i in range(0, len(preds), args.batch):
        m.update(preds[i : i + args.batch], target[i : i + args.batch])
    t_up = time.perf_counter() - t0
    t0 = time.perf_counter()
    res = m.compute()
    t_cmp = time.perf_counter() - t0
    which = "reference-legacy-cpu" if args.reference else "metrics_amd"
    print(f'{{"bench": "map_coco_shape", "framework": "{which}", "images": {args.images}, '
          f'"update_s": {t_up:.3f}, "compute_s": {t_cmp:.3f}, "total_s": {t_up + t_cmp:.3f}, '
          f'"map": {float(res["map"]):.5f}}}')


if __name__ == "__main__":
    main()
