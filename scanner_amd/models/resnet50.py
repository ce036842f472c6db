"""ResNet-50 model family support: weight generation, tensor-file export
for the C++ engine, and a PyTorch fp32 reference forward used by the
numerics tests (tests/test_resnet_gpu.py compares the engine's bf16 MFMA
path against this at fp32).

Layer naming and the [out][r][s][in] (OHWI) weight layout mirror
csrc/ops/resnet50.cpp exactly; BN is pre-folded into per-channel
scale/bias."""
import struct

import numpy as np

BLOCKS = []
for (n, in_c, mid, out, stride) in [(3, 64, 64, 256, 1),
                                    (4, 256, 128, 512, 2),
                                    (6, 512, 256, 1024, 2),
                                    (3, 1024, 512, 2048, 2)]:
    for i in range(n):
        BLOCKS.append({
            "in_c": in_c if i == 0 else out,
            "mid": mid, "out": out,
            "stride": stride if i == 0 else 1,
            "downsample": i == 0,
        })


def conv_specs():
    specs = [("conv1", 3, 64, 7, 2, 3, True)]
    for b, bk in enumerate(BLOCKS):
        p = f"block{b}"
        specs.append((f"{p}.conv1", bk["in_c"], bk["mid"], 1, 1, 0, True))
        specs.append((f"{p}.conv2", bk["mid"], bk["mid"], 3, bk["stride"],
                      1, True))
        specs.append((f"{p}.conv3", bk["mid"], bk["out"], 1, 1, 0, True))
        if bk["downsample"]:
            specs.append((f"{p}.downsample", bk["in_c"], bk["out"], 1,
                          bk["stride"], 0, False))
    specs.append(("fc", 2048, 1000, 1, 1, 0, False))
    return specs


def generate_weights(seed=0):
    """Random He-init weights + folded-BN scale/bias, in OHWI layout."""
    rng = np.random.RandomState(seed)
    ts = {}
    for (name, in_c, out_c, k, stride, pad, relu) in conv_specs():
        fan_in = in_c * k * k
        w = rng.normal(0, np.sqrt(2.0 / fan_in),
                       size=(out_c, k, k, in_c)).astype(np.float32)
        ts[name + ".weight"] = w
        if name == "fc":
            ts[name + ".scale"] = np.ones(out_c, np.float32)
        else:
            ts[name + ".scale"] = rng.uniform(
                0.7, 1.3, out_c).astype(np.float32)
        ts[name + ".bias"] = rng.normal(0, 0.05, out_c).astype(np.float32)
    return ts


def write_tensor_file(path, tensors):
    """BinWriter-compatible tensor file (csrc/ops/resnet50.cpp
    load_tensor_file)."""
    with open(path, "wb") as f:
        f.write(struct.pack("<II", 0x52534E54, len(tensors)))
        for name, arr in tensors.items():
            data = np.ascontiguousarray(arr, np.float32)
            nb = name.encode()
            f.write(struct.pack("<Q", len(nb)))
            f.write(nb)
            f.write(struct.pack("<Q", data.size))
            f.write(data.tobytes())


def torch_reference(tensors, frames_u8, tap=None):
    """fp32 reference forward on CPU via torch.nn.functional. frames_u8:
    [N,H,W,C] u8. Returns [N,1000] f32 logits — or, with `tap` set
    ("conv1", "maxpool", "block<N>", "avgpool"), that point's activation
    as NHWC f32 (matching the op's debug_tap output layout)."""
    import torch
    import torch.nn.functional as F

    def nhwc(t):
        return t.permute(0, 2, 3, 1).contiguous().numpy()

    mean = torch.tensor([0.485, 0.456, 0.406]).view(1, 3, 1, 1)
    std = torch.tensor([0.229, 0.224, 0.225]).view(1, 3, 1, 1)
    x = torch.from_numpy(frames_u8).float().permute(0, 3, 1, 2)  # NCHW
    x = F.interpolate(x, size=(224, 224), mode="bilinear",
                      align_corners=False)
    x = (x / 255.0 - mean) / std

    def conv(name, x, stride, pad, relu, residual=None):
        w = torch.from_numpy(
            np.ascontiguousarray(tensors[name + ".weight"]))
        w = w.permute(0, 3, 1, 2)  # OHWI -> OIHW
        y = F.conv2d(x, w, stride=stride, padding=pad)
        sc = torch.from_numpy(tensors[name + ".scale"]).view(1, -1, 1, 1)
        bi = torch.from_numpy(tensors[name + ".bias"]).view(1, -1, 1, 1)
        y = y * sc + bi
        if residual is not None:
            y = y + residual
        if relu:
            y = F.relu(y)
        return y

    x = conv("conv1", x, 2, 3, True)
    if tap == "conv1":
        return nhwc(x)
    x = F.max_pool2d(x, 3, stride=2, padding=1)
    if tap == "maxpool":
        return nhwc(x)
    for b, bk in enumerate(BLOCKS):
        p = f"block{b}"
        identity = x
        if bk["downsample"]:
            identity = conv(f"{p}.downsample", x, bk["stride"], 0, False)
        y = conv(f"{p}.conv1", x, 1, 0, True)
        y = conv(f"{p}.conv2", y, bk["stride"], 1, True)
        x = conv(f"{p}.conv3", y, 1, 0, True, residual=identity)
        if tap == p:
            return nhwc(x)
    x = x.mean(dim=(2, 3))
    if tap == "avgpool":
        return x.numpy().reshape(x.shape[0], 1, 1, 2048)
    if tap is not None:
        raise ValueError(f"unknown tap '{tap}'")
    w = torch.from_numpy(
        np.ascontiguousarray(tensors["fc.weight"])).view(1000, 2048)
    b = torch.from_numpy(tensors["fc.bias"])
    logits = x @ w.t() + b
    return logits.numpy()
