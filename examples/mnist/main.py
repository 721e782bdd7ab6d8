"""MNIST-style end-to-end example: write a dataset with an image codec,
train a small MLP through the framework's loaders.

Parity role: /root/reference/examples/mnist/ (pytorch_example.py with its
per-row normalize TransformSpec, examples/mnist/pytorch_example.py:92-106).
Runs on CPU or GPU; random data stands in for MNIST (no network access).

Run:  python examples/mnist/main.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..', '..'))

import tempfile

import numpy as np
import torch

from petastorm_amd import TransformSpec, make_reader
from petastorm_amd.codecs import CompressedImageCodec, ScalarCodec
from petastorm_amd.etl.dataset_metadata import materialize_dataset
from petastorm_amd.pytorch import DataLoader
from petastorm_amd.unischema import Unischema, UnischemaField

MnistSchema = Unischema('MnistSchema', [
    UnischemaField('idx', np.int64, (), ScalarCodec(), False),
    UnischemaField('digit', np.int64, (), ScalarCodec(), False),
    UnischemaField('image', np.uint8, (28, 28), CompressedImageCodec('png'),
                   False),
])


def write_dataset(url, rows=600):
    rng = np.random.RandomState(0)
    with materialize_dataset(url, MnistSchema, rowgroup_size_mb=1) as w:
        for i in range(rows):
            w.write_row({'idx': np.int64(i),
                         'digit': np.int64(rng.randint(0, 10)),
                         'image': rng.randint(0, 255, (28, 28),
                                              dtype=np.uint8)})


def train(url, epochs=2, batch_size=64):
    # per-row normalize, like the reference mnist TransformSpec
    def row_transform(row):
        row['image'] = ((row['image'] / 255.0) - 0.1307) / 0.3081
        return row

    ts = TransformSpec(row_transform)
    model = torch.nn.Sequential(torch.nn.Flatten(), torch.nn.Linear(784, 128),
                                torch.nn.ReLU(), torch.nn.Linear(128, 10))
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    for epoch in range(epochs):
        reader = make_reader(url, transform_spec=ts, num_epochs=1,
                             shuffle_row_groups=True, seed=epoch)
        with DataLoader(reader, batch_size=batch_size,
                        shuffling_queue_capacity=256) as loader:
            total, n = 0.0, 0
            for batch in loader:
                x = batch['image'].float()
                y = batch['digit']
                loss = torch.nn.functional.cross_entropy(model(x), y)
                opt.zero_grad()
                loss.backward()
                opt.step()
                total += loss.item()
                n += 1
            print('epoch {} mean loss {:.4f}'.format(epoch, total / n))


if __name__ == '__main__':
    url = 'file://' + tempfile.mkdtemp(prefix='mnist_')
    write_dataset(url)
    train(url)
