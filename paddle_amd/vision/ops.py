"""paddle.vision.ops (reference: python/paddle/vision/ops.py -- detection
ops).  Box/NMS/RoI ops implemented in tensor form on the torch substrate;
codec- and anchor-generator-heavy ops (yolo, deform conv, proposal
generation) are gated until round 2.
"""
from __future__ import annotations

import torch

from ..nn.layer import Layer


def nms(boxes, iou_threshold=0.3, scores=None, category_idxs=None,
        categories=None, top_k=None):
    """Hard NMS (reference: ops.py nms -> phi nms kernel).  boxes [N,4]
    xyxy; returns kept indices sorted by score."""
    n = boxes.shape[0]
    if n == 0:
        return torch.empty(0, dtype=torch.int64, device=boxes.device)
    if scores is None:
        scores = torch.arange(n, 0, -1, dtype=torch.float32,
                              device=boxes.device)
    if category_idxs is not None:
        # category-aware: offset boxes per class so they never overlap
        offs = category_idxs.to(boxes.dtype) * (boxes.max() + 1)
        boxes = boxes + offs.unsqueeze(1)
    order = scores.argsort(descending=True)
    keep = []
    x1, y1, x2, y2 = boxes.unbind(1)
    areas = (x2 - x1).clamp(min=0) * (y2 - y1).clamp(min=0)
    while order.numel() > 0:
        i = order[0]
        keep.append(int(i))
        if order.numel() == 1:
            break
        rest = order[1:]
        xx1 = torch.maximum(x1[i], x1[rest])
        yy1 = torch.maximum(y1[i], y1[rest])
        xx2 = torch.minimum(x2[i], x2[rest])
        yy2 = torch.minimum(y2[i], y2[rest])
        inter = (xx2 - xx1).clamp(min=0) * (yy2 - yy1).clamp(min=0)
        iou = inter / (areas[i] + areas[rest] - inter).clamp(min=1e-10)
        order = rest[iou <= iou_threshold]
    keep_t = torch.tensor(keep, dtype=torch.int64, device=boxes.device)
    if top_k is not None:
        keep_t = keep_t[:top_k]
    return keep_t


def box_coder(prior_box, prior_box_var, target_box,
              code_type="encode_center_size", box_normalized=True, axis=0,
              name=None):
    """Encode/decode boxes against priors (reference: box_coder op)."""
    norm = 0.0 if box_normalized else 1.0
    pw = prior_box[:, 2] - prior_box[:, 0] + norm
    ph = prior_box[:, 3] - prior_box[:, 1] + norm
    pcx = prior_box[:, 0] + pw * 0.5
    pcy = prior_box[:, 1] + ph * 0.5
    if isinstance(prior_box_var, (list, tuple)):
        var = torch.tensor(prior_box_var, dtype=prior_box.dtype,
                           device=prior_box.device).expand(prior_box.shape[0], 4)
    else:
        var = prior_box_var
    if code_type == "encode_center_size":
        tw = target_box[:, 2] - target_box[:, 0] + norm
        th = target_box[:, 3] - target_box[:, 1] + norm
        tcx = target_box[:, 0] + tw * 0.5
        tcy = target_box[:, 1] + th * 0.5
        out = torch.stack([(tcx - pcx) / pw, (tcy - pcy) / ph,
                           torch.log(tw / pw), torch.log(th / ph)], dim=1)
        return out / var if var is not None else out
    # decode_center_size
    t = target_box * var if var is not None else target_box
    cx = t[..., 0] * pw + pcx
    cy = t[..., 1] * ph + pcy
    w = torch.exp(t[..., 2]) * pw
    h = torch.exp(t[..., 3]) * ph
    return torch.stack([cx - w * 0.5, cy - h * 0.5,
                        cx + w * 0.5 - norm, cy + h * 0.5 - norm], dim=-1)


def prior_box(input, image, min_sizes, max_sizes=None, aspect_ratios=(1.0,),
              variance=(0.1, 0.1, 0.2, 0.2), flip=False, clip=False,
              steps=(0.0, 0.0), offset=0.5, min_max_aspect_ratios_order=False,
              name=None):
    """SSD prior boxes over the feature map grid (reference: prior_box op)."""
    fh, fw = input.shape[-2], input.shape[-1]
    ih, iw = image.shape[-2], image.shape[-1]
    step_w = steps[0] or iw / fw
    step_h = steps[1] or ih / fh
    ars = list(aspect_ratios)
    if flip:
        ars += [1.0 / a for a in aspect_ratios if a != 1.0]
    boxes = []
    for y in range(fh):
        for x in range(fw):
            cx = (x + offset) * step_w
            cy = (y + offset) * step_h
            cell = []
            for k, ms in enumerate(min_sizes):
                for ar in ars:
                    w = ms * (ar ** 0.5)
                    h = ms / (ar ** 0.5)
                    cell.append([(cx - w / 2) / iw, (cy - h / 2) / ih,
                                 (cx + w / 2) / iw, (cy + h / 2) / ih])
                if max_sizes:
                    s = (ms * max_sizes[k]) ** 0.5
                    cell.append([(cx - s / 2) / iw, (cy - s / 2) / ih,
                                 (cx + s / 2) / iw, (cy + s / 2) / ih])
            boxes.append(cell)
    out = torch.tensor(boxes, dtype=torch.float32).reshape(fh, fw, -1, 4)
    if clip:
        out = out.clamp(0, 1)
    var = torch.tensor(variance, dtype=torch.float32).expand_as(out)
    return out, var.contiguous()


def roi_pool(x, boxes, boxes_num, output_size, spatial_scale=1.0, name=None):
    """Max-pool each RoI to output_size (reference: roi_pool op)."""
    if isinstance(output_size, int):
        output_size = (output_size, output_size)
    outs = []
    bi = 0
    for b, n in enumerate(boxes_num.tolist()):
        for r in range(n):
            x1, y1, x2, y2 = (boxes[bi + r] * spatial_scale).tolist()
            x1, y1 = int(x1), int(y1)
            x2, y2 = max(int(x2) + 1, x1 + 1), max(int(y2) + 1, y1 + 1)
            patch = x[b:b + 1, :, y1:y2, x1:x2]
            outs.append(torch.nn.functional.adaptive_max_pool2d(
                patch, output_size)[0])
        bi += n
    return torch.stack(outs) if outs else x.new_empty(0, x.shape[1], *output_size)


def roi_align(x, boxes, boxes_num, output_size, spatial_scale=1.0,
              sampling_ratio=-1, aligned=True, name=None):
    """Bilinear RoI align via grid_sample (reference: roi_align op)."""
    if isinstance(output_size, int):
        output_size = (output_size, output_size)
    oh, ow = output_size
    H, W = x.shape[-2], x.shape[-1]
    outs = []
    bi = 0
    half = 0.5 if aligned else 0.0
    for b, n in enumerate(boxes_num.tolist()):
        for r in range(n):
            x1, y1, x2, y2 = (boxes[bi + r] * spatial_scale)
            ys = torch.linspace(float(y1) + half, float(y2) - half, oh,
                                device=x.device)
            xs = torch.linspace(float(x1) + half, float(x2) - half, ow,
                                device=x.device)
            gy = (ys / (H - 1) * 2 - 1).reshape(-1, 1).expand(oh, ow)
            gx = (xs / (W - 1) * 2 - 1).reshape(1, -1).expand(oh, ow)
            grid = torch.stack([gx, gy], dim=-1).unsqueeze(0)
            outs.append(torch.nn.functional.grid_sample(
                x[b:b + 1], grid, align_corners=True)[0])
        bi += n
    return torch.stack(outs) if outs else x.new_empty(0, x.shape[1], oh, ow)


class RoIPool(Layer):
    def __init__(self, output_size, spatial_scale=1.0):
        super().__init__()
        self.output_size = output_size
        self.spatial_scale = spatial_scale

    def forward(self, x, boxes, boxes_num):
        return roi_pool(x, boxes, boxes_num, self.output_size,
                        self.spatial_scale)


class RoIAlign(Layer):
    def __init__(self, output_size, spatial_scale=1.0):
        super().__init__()
        self.output_size = output_size
        self.spatial_scale = spatial_scale

    def forward(self, x, boxes, boxes_num):
        return roi_align(x, boxes, boxes_num, self.output_size,
                         self.spatial_scale)


def distribute_fpn_proposals(fpn_rois, min_level, max_level, refer_level,
                             refer_scale, pixel_offset=False, rois_num=None,
                             name=None):
    """Assign RoIs to FPN levels by scale (reference: ops.py)."""
    off = 1.0 if pixel_offset else 0.0
    w = fpn_rois[:, 2] - fpn_rois[:, 0] + off
    h = fpn_rois[:, 3] - fpn_rois[:, 1] + off
    scale = (w * h).clamp(min=1e-6).sqrt()
    lvl = torch.floor(torch.log2(scale / refer_scale + 1e-8)) + refer_level
    lvl = lvl.clamp(min_level, max_level).long()
    outs, idxs, nums = [], [], []
    for level in range(min_level, max_level + 1):
        m = (lvl == level).nonzero(as_tuple=True)[0]
        outs.append(fpn_rois[m])
        idxs.append(m)
        nums.append(torch.tensor([m.numel()]))
    restore = torch.cat(idxs).argsort() if idxs else None
    return outs, restore, nums


def read_file(filename, name=None):
    with open(filename, "rb") as f:
        data = f.read()
    return torch.frombuffer(bytearray(data), dtype=torch.uint8)


def _gated(name, why):
    def f(*a, **kw):
        raise NotImplementedError(f"{name}: {why}")
    f.__name__ = name
    return f


yolo_loss = _gated("yolo_loss", "YOLO head: round 2")
yolo_box = _gated("yolo_box", "YOLO head: round 2")
generate_proposals = _gated("generate_proposals", "RPN proposal op: round 2")
psroi_pool = _gated("psroi_pool", "position-sensitive RoI pool: round 2")
matrix_nms = _gated("matrix_nms", "soft/matrix NMS: round 2")
decode_jpeg = _gated("decode_jpeg", "JPEG codec not in this image (no PIL); "
                     "decode offline and feed arrays")
deform_conv2d = _gated("deform_conv2d", "deformable conv kernel: round 2")


class PSRoIPool(Layer):
    def __init__(self, *a, **kw):
        super().__init__()
        raise NotImplementedError("psroi_pool: round 2")


class DeformConv2D(Layer):
    def __init__(self, *a, **kw):
        super().__init__()
        raise NotImplementedError("deform_conv2d: round 2")
