// Native COCO greedy matcher (CPU).
//
// Replaces the per-(image,class,area) Python loop of the mAP evaluator with
// ONE call per class: all images' IoU matrices arrive packed in contiguous
// buffers, matching runs for every area range and IoU threshold, outputs are
// bitpacked det-matched / det-ignored flags plus per-(image,area) valid-gt
// counts. (The reference delegates this to pycocotools' C/Python mix;
// spec: torchmetrics detection/_mean_ap.py greedy matching semantics.)
//
// Build: g++ -O3 -shared -fPIC (csrc/build.py) — no GPU involvement; the IoU
// matrices themselves can come from the HIP kernel or host.

#include <algorithm>
#include <cstdint>
#include <cstring>
#include <vector>

extern "C" {

// For one class over n_imgs images:
//  ious:        packed row-major per image [n_dt[i] x n_gt[i]]
//  iou_off:     per-image offset into ious
//  dt_off/gt_off: per-image offsets into det/gt arrays (prefix sums, len n_imgs+1)
//  det_area:    packed det areas (score-desc order, already capped at max_det)
//  gt_crowd:    packed crowd flags (uint8)
//  gt_area:     packed gt areas
//  area_rngs:   [n_areas][2] (lo, hi)
//  iou_thrs:    [n_thrs]
// Outputs (caller-allocated):
//  dtm: [n_areas][n_thrs][total_dt] uint8   (det matched)
//  dti: [n_areas][n_thrs][total_dt] uint8   (det ignored)
//  npig: [n_areas][n_imgs] int32            (non-ignored gt count)
int coco_match_class(
    const float* ious, const int64_t* iou_off,
    const int64_t* dt_off, const int64_t* gt_off, int64_t n_imgs,
    const float* det_area, const uint8_t* gt_crowd, const float* gt_area,
    const float* area_rngs, int64_t n_areas,
    const float* iou_thrs, int64_t n_thrs,
    uint8_t* dtm, uint8_t* dti, int32_t* npig) {
    const int64_t total_dt = dt_off[n_imgs];

    // every (area, image) cell writes disjoint slices of dtm/dti/npig
#pragma omp parallel for collapse(2) schedule(dynamic, 16)
    for (int64_t a = 0; a < n_areas; a++) {
        for (int64_t img = 0; img < n_imgs; img++) {
            const float lo = area_rngs[2 * a], hi = area_rngs[2 * a + 1];
            uint8_t* dtm_a = dtm + a * n_thrs * total_dt;
            uint8_t* dti_a = dti + a * n_thrs * total_dt;
            std::vector<int> gt_order;
            std::vector<uint8_t> gt_ign;
            std::vector<uint8_t> gtm;
            const int64_t d0 = dt_off[img], d1 = dt_off[img + 1];
            const int64_t g0 = gt_off[img], g1 = gt_off[img + 1];
            const int n_dt = (int)(d1 - d0);
            const int n_gt = (int)(g1 - g0);
            const float* iou_img = ious + iou_off[img];

            // per-area gt ignore flags; sort gts ignore-last (stable)
            gt_ign.resize(n_gt);
            gt_order.resize(n_gt);
            int n_keep = 0;
            for (int g = 0; g < n_gt; g++) {
                const float ga = gt_area[g0 + g];
                gt_ign[g] = gt_crowd[g0 + g] || ga < lo || ga > hi;
                if (!gt_ign[g]) n_keep++;
            }
            npig[a * n_imgs + img] = n_keep;
            int k = 0, k2 = n_keep;
            for (int g = 0; g < n_gt; g++) {
                if (!gt_ign[g]) gt_order[k++] = g;
                else gt_order[k2++] = g;
            }

            if (n_dt == 0) continue;
            if (n_gt == 0) {
                // unmatched dets outside the area range are ignored
                for (int64_t t = 0; t < n_thrs; t++) {
                    uint8_t* di = dti_a + t * total_dt + d0;
                    for (int d = 0; d < n_dt; d++) {
                        const float da = det_area[d0 + d];
                        di[d] = (da < lo || da > hi) ? 1 : 0;
                    }
                }
                continue;
            }

            gtm.assign((size_t)n_gt, 0);
            for (int64_t t = 0; t < n_thrs; t++) {
                std::fill(gtm.begin(), gtm.end(), 0);
                const float thr = iou_thrs[t];
                uint8_t* dm = dtm_a + t * total_dt + d0;
                uint8_t* di = dti_a + t * total_dt + d0;
                for (int d = 0; d < n_dt; d++) {
                    float best = thr < (1 - 1e-10f) ? thr : (1 - 1e-10f);
                    int m = -1;
                    for (int gi = 0; gi < n_gt; gi++) {
                        const int g = gt_order[gi];
                        if (gtm[g] && !gt_crowd[g0 + g]) continue;
                        // gts sorted non-ignored first: past them with a match, stop
                        if (m > -1 && !gt_ign[(size_t)m] && gt_ign[g]) break;
                        const float v = iou_img[(int64_t)d * n_gt + g];
                        if (v < best) continue;
                        best = v;
                        m = g;
                    }
                    if (m == -1) {
                        const float da = det_area[d0 + d];
                        if (da < lo || da > hi) di[d] = 1;
                        continue;
                    }
                    di[d] = gt_ign[(size_t)m];
                    dm[d] = 1;
                    gtm[(size_t)m] = 1;
                }
            }
        }
    }
    return 0;
}

// packed all-pairs IoU for one class across images (row-major per image),
// with crowd gts using intersection/det-area.
int coco_iou_class(
    const float* det_boxes, const int64_t* dt_off,
    const float* gt_boxes, const int64_t* gt_off, int64_t n_imgs,
    const uint8_t* gt_crowd, float* ious, const int64_t* iou_off) {
#pragma omp parallel for schedule(dynamic, 32)
    for (int64_t img = 0; img < n_imgs; img++) {
        const int64_t d0 = dt_off[img], d1 = dt_off[img + 1];
        const int64_t g0 = gt_off[img], g1 = gt_off[img + 1];
        const int n_dt = (int)(d1 - d0);
        const int n_gt = (int)(g1 - g0);
        float* out = ious + iou_off[img];
        for (int d = 0; d < n_dt; d++) {
            const float* db = det_boxes + 4 * (d0 + d);
            const float da = (db[2] - db[0]) * (db[3] - db[1]);
            for (int g = 0; g < n_gt; g++) {
                const float* gb = gt_boxes + 4 * (g0 + g);
                const float ga = (gb[2] - gb[0]) * (gb[3] - gb[1]);
                const float ix1 = std::max(db[0], gb[0]);
                const float iy1 = std::max(db[1], gb[1]);
                const float ix2 = std::min(db[2], gb[2]);
                const float iy2 = std::min(db[3], gb[3]);
                const float iw = std::max(ix2 - ix1, 0.0f);
                const float ih = std::max(iy2 - iy1, 0.0f);
                const float inter = iw * ih;
                float uni = gt_crowd[g0 + g] ? da : (da + ga - inter);
                out[(int64_t)d * n_gt + g] = uni > 0 ? inter / uni : 0.0f;
            }
        }
    }
    return 0;
}

}  // extern "C"
