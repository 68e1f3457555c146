"""Self-contained LPIPS perceptual distance.

The reference vendors the PerceptualSimilarity stack on top of torchvision
backbones with bundled linear-head weights
(ESR:loss/PerceptualSimilarity/models/networks_basic.py:32-101,
ESR:loss/restore.py:10-39).  This environment has neither torchvision nor
network access, so the backbones (AlexNet / VGG16 feature stacks) are
implemented here directly.

Weight status:
  * the LEARNED LINEAR HEADS ship with this repo (esr_amd/loss/weights/
    {alex,vgg}.pth — the reference's v0.1 head weights,
    ESR:loss/PerceptualSimilarity/models/weights/v0.1) and load by default;
  * the BACKBONE conv weights still need an ImageNet checkpoint
    (torchvision layout, pass via ``backbone_path``).  Without one the
    backbone is a fixed seeded random projection: deterministic and usable
    for relative comparisons, but NOT paper-comparable — callers can check
    ``LPIPS.backbone_pretrained`` and annotate their outputs.
"""

from __future__ import annotations

from pathlib import Path

import torch
import torch.nn as nn

__all__ = ["PerceptualLoss", "LPIPS"]

_BUNDLED_HEADS = {
    "alex": Path(__file__).resolve().parent / "weights" / "alex.pth",
    "vgg": Path(__file__).resolve().parent / "weights" / "vgg.pth",
}


class _AlexFeatures(nn.Module):
    """AlexNet conv stack split into the 5 LPIPS slices."""

    CH = [64, 192, 384, 256, 256]

    def __init__(self):
        super().__init__()
        self.slice1 = nn.Sequential(
            nn.Conv2d(3, 64, 11, 4, 2), nn.ReLU(inplace=True))
        self.slice2 = nn.Sequential(
            nn.MaxPool2d(3, 2), nn.Conv2d(64, 192, 5, 1, 2), nn.ReLU(inplace=True))
        self.slice3 = nn.Sequential(
            nn.MaxPool2d(3, 2), nn.Conv2d(192, 384, 3, 1, 1), nn.ReLU(inplace=True))
        self.slice4 = nn.Sequential(
            nn.Conv2d(384, 256, 3, 1, 1), nn.ReLU(inplace=True))
        self.slice5 = nn.Sequential(
            nn.Conv2d(256, 256, 3, 1, 1), nn.ReLU(inplace=True))

    def forward(self, x):
        outs = []
        for s in (self.slice1, self.slice2, self.slice3, self.slice4, self.slice5):
            x = s(x)
            outs.append(x)
        return outs


class _VGG16Features(nn.Module):
    """VGG16 conv stack split into the 5 LPIPS slices (relu1_2..relu5_3)."""

    CH = [64, 128, 256, 512, 512]

    @staticmethod
    def _block(cin, cout, n, pool):
        layers = [nn.MaxPool2d(2, 2)] if pool else []
        for i in range(n):
            layers += [nn.Conv2d(cin if i == 0 else cout, cout, 3, 1, 1),
                       nn.ReLU(inplace=True)]
        return nn.Sequential(*layers)

    def __init__(self):
        super().__init__()
        self.slice1 = self._block(3, 64, 2, pool=False)
        self.slice2 = self._block(64, 128, 2, pool=True)
        self.slice3 = self._block(128, 256, 3, pool=True)
        self.slice4 = self._block(256, 512, 3, pool=True)
        self.slice5 = self._block(512, 512, 3, pool=True)

    def forward(self, x):
        outs = []
        for s in (self.slice1, self.slice2, self.slice3, self.slice4, self.slice5):
            x = s(x)
            outs.append(x)
        return outs


def _map_torchvision_backbone(net, sd):
    """Map a torchvision ``alexnet``/``vgg16`` ``features.N.*`` state dict
    onto the slice layout above (conv layers zipped in order)."""
    idxs = sorted({int(k.split(".")[1]) for k in sd
                   if k.startswith("features.") and k.endswith(".weight")})
    convs = [(sd[f"features.{i}.weight"], sd.get(f"features.{i}.bias"))
             for i in idxs]
    out, n = {}, 0
    proto = {"alex": _AlexFeatures, "vgg": _VGG16Features}[net]()
    for name, mod in proto.named_modules():
        if isinstance(mod, nn.Conv2d):
            w, b = convs[n]
            out[f"{name}.weight"] = w
            if b is not None:
                out[f"{name}.bias"] = b
            n += 1
    assert n == len(convs), f"backbone conv count mismatch: {n} vs {len(convs)}"
    return out


def _normalize_tensor(x, eps=1e-10):
    norm = torch.sqrt(torch.sum(x ** 2, dim=1, keepdim=True))
    return x / (norm + eps)


class LPIPS(nn.Module):
    """Learned perceptual distance: unit-normalized feature diffs weighted by
    1x1 linear heads, spatially averaged, summed over slices."""

    # ImageNet normalization applied after the [-1,1] shift, like the
    # original LPIPS 'scaling layer'.
    SHIFT = [-0.030, -0.088, -0.188]
    SCALE = [0.458, 0.448, 0.450]

    def __init__(self, net: str = "alex", weights_path: str | None = None,
                 seed: int = 1234, backbone_path: str | None = None):
        super().__init__()
        # local generator + fork_rng: constructing a metric must neither
        # reseed nor advance the global torch RNG (advisor finding r1);
        # nn.Conv2d default init would otherwise consume global draws
        gen = torch.Generator().manual_seed(seed)
        with torch.random.fork_rng(devices=[]):
            self.features = {"alex": _AlexFeatures,
                             "vgg": _VGG16Features}[net]()
            self.lins = nn.ModuleList([
                nn.Conv2d(c, 1, 1, bias=False) for c in self.features.CH])
        self.backbone_pretrained = False
        with torch.no_grad():
            for p in self.features.parameters():
                if p.dim() > 1:
                    # kaiming-uniform equivalent drawn from the local gen
                    fan_in = p[0].numel()
                    bound = (6.0 / fan_in) ** 0.5
                    p.uniform_(-bound, bound, generator=gen)
                else:
                    p.zero_()
        if backbone_path:
            self.features.load_state_dict(
                _map_torchvision_backbone(
                    net, torch.load(backbone_path, map_location="cpu")))
            self.backbone_pretrained = True
        head_sd = None
        if weights_path:
            head_sd = torch.load(weights_path, map_location="cpu")
        elif _BUNDLED_HEADS.get(net, Path("/nonexistent")).exists():
            # the reference's trained v0.1 linear heads, bundled as data
            head_sd = torch.load(str(_BUNDLED_HEADS[net]), map_location="cpu")
        if head_sd is not None:
            self._load_heads(head_sd)
        else:
            for lin in self.lins:
                with torch.no_grad():
                    lin.weight.uniform_(0.0, 0.1, generator=gen)
        self.heads_pretrained = head_sd is not None
        self.register_buffer("shift", torch.tensor(self.SHIFT).view(1, 3, 1, 1))
        self.register_buffer("scale", torch.tensor(self.SCALE).view(1, 3, 1, 1))
        for p in self.parameters():
            p.requires_grad_(False)
        self.eval()

    def _load_heads(self, sd):
        """Load linear-head weights from either this module's own layout
        (``lins.N.weight``) or the reference's
        (``linN.model.1.weight`` — ESR:loss/PerceptualSimilarity/models/
        networks_basic.py:32-101)."""
        with torch.no_grad():
            for i, lin in enumerate(self.lins):
                for key in (f"lins.{i}.weight", f"lin{i}.model.1.weight"):
                    if key in sd:
                        lin.weight.copy_(sd[key])
                        break
                else:
                    raise KeyError(f"no head weight for slice {i} in state dict")

    def forward(self, pred, target, normalize=True):
        if normalize:  # [0,1] -> [-1,1]
            pred = 2 * pred - 1
            target = 2 * target - 1
        pred = (pred - self.shift) / self.scale
        target = (target - self.shift) / self.scale
        f0 = self.features(pred)
        f1 = self.features(target)
        dist = 0
        for k, lin in enumerate(self.lins):
            d = (_normalize_tensor(f0[k]) - _normalize_tensor(f1[k])) ** 2
            dist = dist + lin(d).mean(dim=(2, 3))
        return dist


class PerceptualLoss:
    """N-channel wrapper (parity: ESR:loss/restore.py:10-39): 1-ch inputs are
    replicated to RGB; >3-ch inputs are scored per channel and averaged."""

    def __init__(self, weight=1.0, net="alex", device="cpu", weights_path=None,
                 backbone_path=None):
        self.model = LPIPS(net=net, weights_path=weights_path,
                           backbone_path=backbone_path).to(device)
        self.weight = weight

    @property
    def paper_comparable(self):
        """True only when both backbone and heads carry trained weights."""
        return self.model.backbone_pretrained and self.model.heads_pretrained

    @torch.no_grad()
    def __call__(self, pred, target, normalize=True):
        assert pred.shape == target.shape
        C = pred.shape[1]
        if C == 1:
            pred = pred.repeat(1, 3, 1, 1)
            target = target.repeat(1, 3, 1, 1)
            dist = self.model(pred, target, normalize)
        elif C == 3:
            dist = self.model(pred, target, normalize)
        else:
            dist = 0
            for idx in range(C):
                dist = dist + self.model(pred[:, idx:idx + 1].repeat(1, 3, 1, 1),
                                         target[:, idx:idx + 1].repeat(1, 3, 1, 1),
                                         normalize)
            dist = dist / C
        return self.weight * dist.mean()
