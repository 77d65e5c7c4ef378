#!/usr/bin/env python3
"""ImageFeaturizer transfer-learning throughput (BASELINE config #4:
ResNet-50 on CIFAR10-shaped 32x32x3, PyTorch-ROCm, DP via DDP/RCCL).
One step = one fine-tune minibatch (forward+backward+optimizer);
value = images/sec aggregate."""
import argparse
import json
import os
import sys
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=512)
    ap.add_argument("--image-size", type=int, default=32)
    ap.add_argument("--bytes-to-features", action="store_true",
                    help="also measure JPEG bytes → ResNet features "
                         "(codec + featurize end to end)")
    args = ap.parse_args()
    if args.bytes_to_features:
        bytes_to_features(args)

    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from mmlspark_amd.models.resnet import resnet50
    from mmlspark_amd.parallel.comm import init_from_env

    comm = init_from_env()
    rank, world = comm.rank, comm.world_size
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda") if use_gpu else torch.device("cpu")
    if use_gpu and world > 1:
        # modulo lets a one-GPU rehearsal run the multi-proc path (gloo)
        li = int(os.environ.get("LOCAL_RANK", rank)) % torch.cuda.device_count()
        torch.cuda.set_device(li)
        device = torch.device("cuda", li)
    bs = args.batch if use_gpu else 16

    torch.manual_seed(7 + rank)
    net = resnet50(num_classes=10).to(device)
    train_net = net
    if comm.is_distributed:
        train_net = torch.nn.parallel.DistributedDataParallel(
            net, device_ids=[device.index] if use_gpu else None)
    opt = torch.optim.SGD(train_net.parameters(), lr=0.1, momentum=0.9)
    X = torch.randn(bs, 3, args.image_size, args.image_size, device=device)
    y = torch.randint(0, 10, (bs,), device=device)

    amp = torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                         enabled=use_gpu)

    def step():
        opt.zero_grad(set_to_none=True)
        with amp:
            loss = torch.nn.functional.cross_entropy(train_net(X), y)
        loss.backward()
        opt.step()

    for _ in range(args.warmup):
        step()
    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.perf_counter() - t0
    t = torch.tensor([elapsed], dtype=torch.float64)
    if comm.is_distributed:
        comm.all_reduce(t.to(device) if use_gpu else t, op="max")
    elapsed = float(t[0])

    if rank == 0:
        print(json.dumps({
            "metric": "image_featurizer_images_per_sec",
            "value": bs * world * args.steps / elapsed,
            "unit": "images/s", "n_gpus": world, "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None, "dtype": "bf16", "data": "synthetic",
            "config": {"model": "ResNet50 fine-tune",
                       "global_batch": bs * world,
                       "image": f"{args.image_size}x{args.image_size}x3",
                       "parallelism": f"dp{world} (DDP over RCCL)"},
        }), flush=True)


def bytes_to_features(args):
    """JPEG bytes → decode (native C++) → ImageFeaturizer forward.
    Measures the full ingestion path the round-1 synthetic-tensor number
    skipped; runs before the training bench and
    exits."""
    import numpy as np
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    import pandas as pd
    from mmlspark_amd.io_http.jpeg_codec import encode_jpeg
    from mmlspark_amd.models.image_featurizer import ImageFeaturizer

    use_gpu = torch.cuda.is_available()
    n_imgs = 2048 if use_gpu else 64
    rng = np.random.default_rng(0)
    side = 224
    yy, xx = np.mgrid[0:side, 0:side]
    base = (np.sin(xx / 9.0) * np.cos(yy / 13.0) * 90 + 128)
    imgs = []
    for i in range(16):  # 16 distinct images cycled
        img = np.clip(np.stack([np.roll(base, i, 0)] * 3, -1)
                      + rng.normal(0, 8, (side, side, 3)), 0, 255)
        imgs.append(encode_jpeg(img.astype(np.uint8), quality=90))
    blobs = [imgs[i % 16] for i in range(n_imgs)]
    feat = ImageFeaturizer(modelName="ResNet50", cutOutputLayers=1,
                           imageSize=224)
    # warm with the SAME batch shape as the measured run — MIOpen tunes
    # per conv shape and the first find is seconds, not milliseconds
    nb = min(len(blobs), 2 * feat.get("batchSize"))
    feat.transform(pd.DataFrame({"image": blobs[:nb]}))
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = feat.transform(pd.DataFrame({"image": blobs}))
    if use_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "image_bytes_to_features_per_sec",
        "value": n_imgs / dt,
        "unit": "images/s",
        "higher_is_better": True,
        "n_images": n_imgs,
        "jpeg": "224x224 q90, native C++ decoder",
        "model": "ResNet50 cutOutputLayers=1",
    }), flush=True)
    sys.exit(0)


if __name__ == "__main__":
    main()
