"""ResNet-50 gradient bucket model for BASELINE config #4.

Parameter tensor sizes of ResNet-50 (v1.5, 1000 classes), partitioned
BytePS-style into <= partition_bytes chunks (the reference's
BYTEPS_PARTITION_BYTES mechanism) — each chunk is one PS
key pushed/pulled per step.
"""


def resnet50_param_sizes():
    """Parameter tensor sizes in elements (fp32), in layer order."""
    sizes = []

    def conv(out_c, in_c, k):
        sizes.append(out_c * in_c * k * k)

    def bn(c):
        sizes.append(c)  # weight
        sizes.append(c)  # bias

    conv(64, 3, 7)
    bn(64)
    # (planes, blocks, in_planes at stage entry)
    stages = [(64, 3, 64), (128, 4, 256), (256, 6, 512), (512, 3, 1024)]
    for planes, blocks, in_planes in stages:
        for b in range(blocks):
            inp = in_planes if b == 0 else planes * 4
            conv(planes, inp, 1)
            bn(planes)
            conv(planes, planes, 3)
            bn(planes)
            conv(planes * 4, planes, 1)
            bn(planes * 4)
            if b == 0:  # downsample
                conv(planes * 4, inp, 1)
                bn(planes * 4)
    sizes.append(1000 * 2048)  # fc weight
    sizes.append(1000)  # fc bias
    return sizes


def resnet50_grad_buckets(partition_bytes=4 << 20, dtype_bytes=4):
    """Byte sizes of the gradient messages (tensors split into
    <= partition_bytes chunks). Sums to ~102 MB fp32."""
    buckets = []
    for n in resnet50_param_sizes():
        nbytes = n * dtype_bytes
        while nbytes > partition_bytes:
            buckets.append(partition_bytes)
            nbytes -= partition_bytes
        if nbytes:
            buckets.append(nbytes)
    return buckets
