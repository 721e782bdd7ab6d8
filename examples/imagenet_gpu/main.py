"""ImageNet-style training-input pipeline on the MI355X GPU decode path.

Parity role: /root/reference/examples/imagenet/ (schema + generator) — but
with the decode running as HIP kernels: Parquet pages, snappy, JPEG Huffman/
IDCT/color and the NHWC->NCHW normalize all execute on the GPU; the training
loop receives ready NCHW fp32 CUDA tensors.

Run on a GPU box:  python examples/imagenet_gpu/main.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import tempfile

import numpy as np
import torch

from petastorm_amd import make_batch_reader, ops
from petastorm_amd.pytorch import BatchedDataLoader
from petastorm_amd.test_util.dataset_gen import create_imagenet_dataset


def main(url=None, rows=512, batch_size=64, steps=20):
    assert torch.cuda.is_available(), 'this example needs a GPU'
    if url is None:
        url = 'file://' + tempfile.mkdtemp(prefix='imnet_')
        create_imagenet_dataset(url, num_rows=rows, rowgroup_size_mb=16)
    elif not url.startswith('file://'):
        import os
        if not os.path.isdir(url) or not os.listdir(url):
            os.makedirs(url, exist_ok=True)
            create_imagenet_dataset('file://' + url, num_rows=rows,
                                    rowgroup_size_mb=16)
        url = 'file://' + url

    ops.ext()  # loud failure if the native extension is missing
    # fused into the jpeg color kernel epilogue: decode emits normalized
    # NCHW float32 directly (no NHWC uint8 intermediate)
    from petastorm_amd.transform import fused_image_normalize
    ts = fused_image_normalize('image', mean=[0.485, 0.456, 0.406],
                               std=[0.229, 0.224, 0.225])

    # a tiny conv net standing in for the real model
    model = torch.nn.Sequential(
        torch.nn.Conv2d(3, 16, 7, stride=4), torch.nn.ReLU(),
        torch.nn.AdaptiveAvgPool2d(1), torch.nn.Flatten(),
        torch.nn.Linear(16, 1000)).cuda()
    opt = torch.optim.SGD(model.parameters(), lr=0.01)

    reader = make_batch_reader(url, device='cuda', num_epochs=None,
                               shuffle_row_groups=True, seed=0,
                               transform_spec=ts)
    loader = BatchedDataLoader(reader, batch_size=batch_size)
    it = iter(loader)
    for step in range(steps):
        batch = next(it)
        logits = model(batch['image'])
        loss = torch.nn.functional.cross_entropy(logits,
                                                 batch['label'].long())
        opt.zero_grad()
        loss.backward()
        opt.step()
        if step % 5 == 0:
            print('step {} loss {:.4f}'.format(step, loss.item()))
    reader.stop()
    reader.join()
    print('done; decode diagnostics:', reader.diagnostics)


if __name__ == '__main__':
    import argparse
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument('url', nargs='?', default=None,
                        help='dataset URL (default: generate a fresh one)')
    main(parser.parse_args().url)
