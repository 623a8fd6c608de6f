"""Detection/vision contrib ops (reference src/operator/contrib/:
bounding_box.cc box_iou/box_nms, multibox_prior.cc, roi_align.cc).

Torch-tensor level; elementwise/sort-based pieces use the library ops,
roi_align is a hand-rolled bilinear gather (differentiable).
"""
import torch

__all__ = ['box_iou', 'box_nms', 'multibox_prior', 'roi_align',
           'bbox_transform']


def box_iou(lhs, rhs, fmt='corner'):
    """IoU matrix [N, M] (reference _contrib_box_iou)."""
    if fmt == 'center':
        lhs = _center_to_corner(lhs)
        rhs = _center_to_corner(rhs)
    lt = torch.maximum(lhs[..., :, None, :2], rhs[..., None, :, :2])
    rb = torch.minimum(lhs[..., :, None, 2:], rhs[..., None, :, 2:])
    wh = (rb - lt).clamp(min=0)
    inter = wh[..., 0] * wh[..., 1]
    area_l = ((lhs[..., 2] - lhs[..., 0]) *
              (lhs[..., 3] - lhs[..., 1]))[..., :, None]
    area_r = ((rhs[..., 2] - rhs[..., 0]) *
              (rhs[..., 3] - rhs[..., 1]))[..., None, :]
    return inter / (area_l + area_r - inter).clamp(min=1e-12)


def _center_to_corner(b):
    xy, wh = b[..., :2], b[..., 2:4]
    half = wh / 2
    return torch.cat([xy - half, xy + half], dim=-1)


def box_nms(data, overlap_thresh=0.5, valid_thresh=0, topk=-1, coord_start=2,
            score_index=1, id_index=0, force_suppress=False):
    """Greedy NMS (reference _contrib_box_nms semantics): data
    [..., N, k] with [id, score, x1, y1, x2, y2]; suppressed entries get
    id/score -1.  Returns the filtered copy, scores sorted descending."""
    shape = data.shape
    out = data.reshape(-1, shape[-2], shape[-1]).clone()
    for b in range(out.shape[0]):
        boxes_all = out[b]
        score = boxes_all[:, score_index]
        order = torch.argsort(score, descending=True)
        boxes_all = boxes_all[order]
        keep_rows = []
        suppressed = torch.zeros(boxes_all.shape[0], dtype=torch.bool,
                                 device=data.device)
        valid = score[order] > valid_thresh
        coords = boxes_all[:, coord_start:coord_start + 4]
        iou = box_iou(coords, coords)
        for i in range(boxes_all.shape[0]):
            if suppressed[i] or not bool(valid[i]):
                continue
            keep_rows.append(i)
            if topk > 0 and len(keep_rows) >= topk:
                suppressed[i + 1:] = True
                break
            same_cls = (force_suppress or id_index < 0 or
                        (boxes_all[:, id_index] == boxes_all[i, id_index]))
            suppressed |= (iou[i] > overlap_thresh) & same_cls
            suppressed[i] = False
        kill = torch.ones(boxes_all.shape[0], dtype=torch.bool,
                          device=data.device)
        kill[torch.tensor(keep_rows, dtype=torch.long,
                          device=data.device)] = False
        boxes_all = boxes_all.clone()
        if id_index >= 0:
            boxes_all[kill, id_index] = -1
        boxes_all[kill, score_index] = -1
        out[b] = boxes_all
    return out.reshape(shape)


def multibox_prior(data, sizes=(1.0,), ratios=(1.0,), clip=False,
                   steps=(-1.0, -1.0), offsets=(0.5, 0.5)):
    """Anchor boxes [1, H*W*(S+R-1), 4] (reference multibox_prior.cc)."""
    H, W = data.shape[-2], data.shape[-1]
    dev = data.device if isinstance(data, torch.Tensor) else None
    step_y = steps[0] if steps[0] > 0 else 1.0 / H
    step_x = steps[1] if steps[1] > 0 else 1.0 / W
    cy = (torch.arange(H, device=dev, dtype=torch.float32) + offsets[0]) * step_y
    cx = (torch.arange(W, device=dev, dtype=torch.float32) + offsets[1]) * step_x
    anchors = []
    for i, s in enumerate(sizes):
        for j, r in enumerate(ratios):
            if i > 0 and j > 0:
                continue  # reference: sizes x first ratio + first size x ratios
            w = s * (r ** 0.5) / 2
            h = s / (r ** 0.5) / 2
            anchors.append((w, h))
    boxes = []
    for yy in cy:
        for xx in cx:
            for w, h in anchors:
                boxes.append([xx - w, yy - h, xx + w, yy + h])
    t = torch.tensor(boxes, device=dev, dtype=torch.float32).unsqueeze(0)
    if clip:
        t = t.clamp(0, 1)
    return t


def roi_align(data, rois, pooled_size, spatial_scale=1.0, sample_ratio=2):
    """ROIAlign (reference roi_align.cc): data [N,C,H,W], rois [R,5]
    (batch_idx, x1, y1, x2, y2); bilinear sampling, average pooled."""
    N, C, H, W = data.shape
    ph, pw = (pooled_size, pooled_size) if isinstance(pooled_size, int) \
        else pooled_size
    R = rois.shape[0]
    out = data.new_zeros(R, C, ph, pw)
    for r in range(R):
        bi = int(rois[r, 0].item())
        x1, y1, x2, y2 = (rois[r, 1:5] * spatial_scale).tolist()
        rw = max(x2 - x1, 1e-3)
        rh = max(y2 - y1, 1e-3)
        # normalized grid for grid_sample over the roi
        ys = torch.linspace(y1, y2, ph * sample_ratio, device=data.device)
        xs = torch.linspace(x1, x2, pw * sample_ratio, device=data.device)
        gy = (ys / max(H - 1, 1)) * 2 - 1
        gx = (xs / max(W - 1, 1)) * 2 - 1
        grid = torch.stack(torch.meshgrid(gy, gx, indexing='ij'),
                           dim=-1)[None, :, :, [1, 0]]
        sampled = torch.nn.functional.grid_sample(
            data[bi:bi + 1].float(), grid.float(), mode='bilinear',
            align_corners=True)
        pooled = torch.nn.functional.avg_pool2d(sampled, sample_ratio)
        out[r] = pooled[0].to(out.dtype)
    return out


def bbox_transform(anchors, deltas, means=(0., 0., 0., 0.),
                   stds=(1., 1., 1., 1.)):
    """Apply box regression deltas (reference bbox utils)."""
    widths = anchors[..., 2] - anchors[..., 0]
    heights = anchors[..., 3] - anchors[..., 1]
    cx = anchors[..., 0] + widths / 2
    cy = anchors[..., 1] + heights / 2
    m = torch.tensor(means, device=deltas.device)
    s = torch.tensor(stds, device=deltas.device)
    d = deltas * s + m
    ncx = cx + d[..., 0] * widths
    ncy = cy + d[..., 1] * heights
    nw = widths * torch.exp(d[..., 2])
    nh = heights * torch.exp(d[..., 3])
    return torch.stack([ncx - nw / 2, ncy - nh / 2,
                        ncx + nw / 2, ncy + nh / 2], dim=-1)
