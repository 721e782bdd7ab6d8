"""TensorFlow adapters: feed a Reader into tf graph-mode tensors or tf.data.

Parity: /root/reference/petastorm/tf_utils.py.

* dtype sanitization: Decimal -> str, datetime -> int64 ns, uint16/uint32
  widened (reference :27-43, :57-96)
* ``tf_tensors(reader)``: graph-mode ``tf.py_func`` wrapping ``next(reader)``
  with an optional ``tf.RandomShuffleQueue`` (reference :269-318); the queue
  size is observable through the well-known op name
  ``random_shuffling_queue_size`` (reference :46-47, :206-209)
* ``make_petastorm_dataset(reader)``: ``tf.data.Dataset.from_generator`` +
  namedtuple map with static shapes (reference :336-405), auto
  ``reader.reset()`` on re-iteration (reference :374-380)
* NGram flatten/unflatten across the py_func boundary (reference :140-182,
  :408-418)

TensorFlow is imported lazily: this module imports fine without TF; calling
any adapter without TF installed raises an informative ImportError.  (The
MI355X serving environment is PyTorch-first; the TF adapter exists for API
parity and runs wherever tensorflow-rocm is present.)
"""

from decimal import Decimal

import numpy as np

_SHUFFLING_QUEUE_SIZE_NAME = 'random_shuffling_queue_size'


def _tf():
    try:
        import tensorflow as tf  # noqa: F811
        return tf
    except ImportError as e:
        raise ImportError(
            'petastorm_amd.tf_utils requires tensorflow (tensorflow-rocm on '
            'this platform); install it or use petastorm_amd.pytorch') from e


_NUMPY_TO_TF_SANITIZED = {
    np.uint16: np.int32,
    np.uint32: np.int64,
}


def _sanitize_field_tf_types(sample):
    """Convert row values TF can't represent (reference tf_utils.py:57-96)."""
    next_sample_dict = sample._asdict()
    for name, value in next_sample_dict.items():
        if value is None:
            raise RuntimeError('Encountered "{}"=None. Tensorflow does not '
                               'support None values as a tensor. Consider '
                               'filtering out None values using a predicate'
                               .format(name))
        if isinstance(value, Decimal):
            next_sample_dict[name] = str(value)
        elif isinstance(value, np.ndarray):
            if value.dtype.type in _NUMPY_TO_TF_SANITIZED:
                next_sample_dict[name] = value.astype(
                    _NUMPY_TO_TF_SANITIZED[value.dtype.type])
            elif np.issubdtype(value.dtype, np.datetime64):
                next_sample_dict[name] = value.astype('datetime64[ns]') \
                    .astype(np.int64)
            elif value.dtype.type == np.object_ and value.size and \
                    isinstance(value.flat[0], Decimal):
                next_sample_dict[name] = np.vectorize(str)(value)
        elif isinstance(value, np.number) and \
                type(value) in _NUMPY_TO_TF_SANITIZED:
            next_sample_dict[name] = _NUMPY_TO_TF_SANITIZED[type(value)](value)
        elif isinstance(value, np.datetime64):
            next_sample_dict[name] = value.astype('datetime64[ns]') \
                .astype(np.int64)
    return sample.__class__(**next_sample_dict)


def _numpy_to_tf_dtype(field):
    """Map a UnischemaField to its TF dtype (reference tf_utils.py:27-43)."""
    tf = _tf()
    np_dtype = field.numpy_dtype
    if np_dtype is Decimal or np_dtype in (np.str_, np.bytes_):
        return tf.string
    if np_dtype is np.datetime64:
        return tf.int64
    np_dtype = np.dtype(np_dtype).type
    if np_dtype in _NUMPY_TO_TF_SANITIZED:
        np_dtype = _NUMPY_TO_TF_SANITIZED[np_dtype]
    return tf.as_dtype(np.dtype(np_dtype))


def _schema_to_tf_dtypes(schema):
    return [_numpy_to_tf_dtype(f) for f in schema.fields.values()]


def _schema_to_tf_dtypes_ngram(schema, ngram):
    """Flattened dtype list over all timesteps (reference :99-120)."""
    dtypes = []
    for ts in sorted(ngram.fields.keys()):
        view = ngram.get_schema_at_timestep(schema, ts)
        dtypes.extend(_schema_to_tf_dtypes(view))
    return dtypes


def _flatten(data):
    """{timestep: namedtuple} -> flat tuple (reference :140-158)."""
    flat = []
    for ts in sorted(data.keys()):
        flat.extend(data[ts])
    return tuple(flat)


def make_namedtuple_tf_ngram(schema, ngram, *args):
    """Rebuild {timestep: namedtuple} from flat tensors (reference :160-182)."""
    out = {}
    idx = 0
    for ts in sorted(ngram.fields.keys()):
        view = ngram.get_schema_at_timestep(schema, ts)
        n = len(view.fields)
        out[ts] = view.make_namedtuple(
            **dict(zip(view.fields.keys(), args[idx:idx + n])))
        idx += n
    return out


def _shuffling_queue(shuffling_queue_capacity, min_after_dequeue, dtypes,
                     fields_as_list):
    """Wrap tensors in a RandomShuffleQueue (reference :201-219)."""
    tf = _tf()
    queue = tf.queue.RandomShuffleQueue(shuffling_queue_capacity,
                                        min_after_dequeue, dtypes)
    enqueue_op = queue.enqueue(fields_as_list)
    queue_runner = tf.compat.v1.train.QueueRunner(queue, [enqueue_op])
    tf.compat.v1.train.add_queue_runner(queue_runner)
    tf.identity(queue.size(), name=_SHUFFLING_QUEUE_SIZE_NAME)
    return queue.dequeue()


def tf_tensors(reader, shuffling_queue_capacity=0, min_after_dequeue=0):
    """Graph-mode tensors over a Reader (reference tf_utils.py:269-318)."""
    tf = _tf()
    if reader.batched_output and shuffling_queue_capacity > 0:
        raise ValueError('shuffling_queue is not supported with batched '
                         'output (reference tf_utils.py:307-311)')

    if reader.ngram is not None:
        dtypes = _schema_to_tf_dtypes_ngram(reader.schema, reader.ngram)

        def gen():
            sample = next(reader)
            sample = {k: _sanitize_field_tf_types(v)
                      for k, v in sample.items()}
            return _flatten(sample)

        fields_as_list = tf.compat.v1.py_func(gen, [], dtypes)
        if shuffling_queue_capacity > 0:
            fields_as_list = _shuffling_queue(shuffling_queue_capacity,
                                              min_after_dequeue, dtypes,
                                              fields_as_list)
        return make_namedtuple_tf_ngram(reader.schema, reader.ngram,
                                        *fields_as_list)

    dtypes = _schema_to_tf_dtypes(reader.schema)

    def gen():
        return tuple(_sanitize_field_tf_types(next(reader)))

    fields_as_list = tf.compat.v1.py_func(gen, [], dtypes)
    if shuffling_queue_capacity > 0:
        fields_as_list = _shuffling_queue(shuffling_queue_capacity,
                                          min_after_dequeue, dtypes,
                                          fields_as_list)
    # restore static shapes where known
    named = []
    for tensor, field in zip(fields_as_list, reader.schema.fields.values()):
        if field.shape is not None and not reader.batched_output and \
                all(d is not None for d in field.shape):
            tensor.set_shape(field.shape)
        named.append(tensor)
    return reader.schema._get_namedtuple()(*named)


def make_petastorm_dataset(reader):
    """tf.data.Dataset over a Reader (reference tf_utils.py:336-405)."""
    tf = _tf()

    def dataset_generator():
        if reader.last_row_consumed:
            # auto-reset on re-iteration (reference :374-380)
            reader.reset()
        for row in reader:
            if reader.ngram is not None:
                yield _flatten({k: _sanitize_field_tf_types(v)
                                for k, v in row.items()})
            else:
                yield tuple(_sanitize_field_tf_types(row))

    if reader.ngram is not None:
        dtypes = tuple(_schema_to_tf_dtypes_ngram(reader.schema,
                                                  reader.ngram))
        dataset = tf.data.Dataset.from_generator(dataset_generator, dtypes)
        return dataset.map(lambda *args: make_namedtuple_tf_ngram(
            reader.schema, reader.ngram, *args))

    dtypes = tuple(_schema_to_tf_dtypes(reader.schema))
    dataset = tf.data.Dataset.from_generator(dataset_generator, dtypes)
    nt = reader.schema._get_namedtuple()

    def set_shapes(*fields):
        out = []
        for tensor, field in zip(fields, reader.schema.fields.values()):
            if field.shape is not None and not reader.batched_output and \
                    all(d is not None for d in field.shape):
                tensor.set_shape(field.shape)
            out.append(tensor)
        return nt(*out)

    return dataset.map(set_shapes)


def date_to_nsec_from_epoch(dt):
    """Seconds-resolution datetime/date -> int64 ns since epoch
    (reference tf_utils.py:50-54)."""
    from calendar import timegm
    return timegm(dt.timetuple()) * 1_000_000_000
