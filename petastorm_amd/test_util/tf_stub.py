"""A minimal, faithful TensorFlow stand-in for exercising tf_utils.

TensorFlow (even CPU) is not installable in this offline environment, so
the TF adapter is executed against this stub: a tiny graph/eager engine
implementing exactly the API surface ``petastorm_amd/tf_utils.py`` touches
(py_func graph nodes + Session.run, RandomShuffleQueue + QueueRunner,
tf.data.Dataset.from_generator/map, dtype objects, set_shape validation).
Semantics mirror real TF closely enough that dtype mapping, NGram
flatten/unflatten, queue shuffling, auto-reset and static-shape bugs in the
adapter are observable.  Reference behaviors being modeled:
/root/reference/petastorm/tf_utils.py:269-318 (tf_tensors),
:336-405 (make_petastorm_dataset), :201-219 (shuffling queue).

This is a TEST HARNESS, not a TF implementation; install tensorflow-rocm
to run the adapter for real.
"""

import random
import types

import numpy as np


class DType(object):
    def __init__(self, name, np_dtype):
        self.name = name
        self.as_numpy_dtype = np_dtype

    def __repr__(self):
        return 'tf.' + self.name

    def __eq__(self, other):
        return isinstance(other, DType) and other.name == self.name

    def __hash__(self):
        return hash(self.name)


string = DType('string', np.object_)
int8 = DType('int8', np.int8)
int16 = DType('int16', np.int16)
int32 = DType('int32', np.int32)
int64 = DType('int64', np.int64)
uint8 = DType('uint8', np.uint8)
float32 = DType('float32', np.float32)
float64 = DType('float64', np.float64)
bool_ = DType('bool', np.bool_)

_DTYPES = {np.dtype(d.as_numpy_dtype): d
           for d in (int8, int16, int32, int64, uint8, float32, float64,
                     bool_)}


def as_dtype(np_dtype):
    d = np.dtype(np_dtype)
    if d.kind in 'SU' or d == np.dtype(object):
        return string
    if d in _DTYPES:
        return _DTYPES[d]
    raise TypeError('tf_stub: unsupported dtype {!r}'.format(np_dtype))


def _check_value(value, dtype, shape, ctx):
    if dtype is string:
        return value
    arr = np.asarray(value)
    want = np.dtype(dtype.as_numpy_dtype)
    if arr.dtype != want:
        raise TypeError('{}: value dtype {} != declared {}'.format(
            ctx, arr.dtype, want))
    if shape is not None:
        if len(arr.shape) != len(shape) or any(
                s is not None and s != a for s, a in zip(shape, arr.shape)):
            raise ValueError('{}: value shape {} incompatible with '
                             'static shape {}'.format(ctx, arr.shape, shape))
    return arr


class Tensor(object):
    """Symbolic graph tensor: an output slot of a node."""

    def __init__(self, node, index, dtype, name=None):
        self._node = node
        self._index = index
        self.dtype = dtype
        self._shape = None
        self.name = name or 'tensor'

    def set_shape(self, shape):
        self._shape = tuple(shape)

    def get_shape(self):
        return self._shape


class _PyFuncNode(object):
    def __init__(self, func, dtypes):
        self.func = func
        self.dtypes = list(dtypes)

    def run(self):
        out = self.func()
        if not isinstance(out, (tuple, list)):
            out = (out,)
        if len(out) != len(self.dtypes):
            raise ValueError('py_func returned {} values, declared {}'
                             .format(len(out), len(self.dtypes)))
        return list(out)


class _IdentityNode(object):
    def __init__(self, source):
        self.source = source  # callable -> value

    def run(self):
        return [self.source()]


class RandomShuffleQueue(object):
    """Functional model of tf.queue.RandomShuffleQueue: enqueue fills from
    the producer node; dequeue blocks until > min_after_dequeue items are
    present, then removes a RANDOM item (reference queue semantics)."""

    def __init__(self, capacity, min_after_dequeue, dtypes, name=None):
        self.capacity = capacity
        self.min_after_dequeue = min_after_dequeue
        self.dtypes = list(dtypes)
        self._items = []
        self._producers = []
        self._rng = random.Random(0)

    def enqueue(self, tensors):
        return ('enqueue', self, tuple(tensors))

    def _fill_once(self, session):
        op = self._producers[0]
        vals = session._evaluate_list(op[2])
        self._items.append(vals)

    def dequeue(self):
        node = _QueueDequeueNode(self)
        return tuple(Tensor(node, i, dt) for i, dt in enumerate(self.dtypes))

    def size(self):
        node = _IdentityNode(lambda: np.int32(len(self._items)))
        return Tensor(node, 0, int32)


class _QueueDequeueNode(object):
    def __init__(self, queue):
        self.queue = queue
        self._session = None

    def run(self):
        q = self.queue
        while len(q._items) <= q.min_after_dequeue or not q._items:
            q._fill_once(self._session)
        idx = q._rng.randrange(len(q._items))
        return list(q._items.pop(idx))


class QueueRunner(object):
    def __init__(self, queue, enqueue_ops):
        self.queue = queue
        for op in enqueue_ops:
            queue._producers.append(op)


_COLLECTED_RUNNERS = []


def _add_queue_runner(runner):
    _COLLECTED_RUNNERS.append(runner)


def identity(tensor, name=None):
    t = Tensor(tensor._node, tensor._index, tensor.dtype, name=name)
    t._shape = tensor._shape
    return t


class Session(object):
    """Evaluates fetches by running each distinct producing node ONCE per
    Session.run call (matching TF-graph py_func semantics)."""

    def __init__(self):
        pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False

    def _evaluate_list(self, tensors):
        cache = {}
        out = []
        for t in tensors:
            node = t._node
            if isinstance(node, _QueueDequeueNode):
                node._session = self
            if id(node) not in cache:
                cache[id(node)] = node.run()
            vals = cache[id(node)]
            out.append(_check_value(vals[t._index], t.dtype, t._shape,
                                    'Session.run'))
        return out

    def run(self, fetches):
        if isinstance(fetches, Tensor):
            return self._evaluate_list([fetches])[0]
        if isinstance(fetches, dict):
            flat = {}
            for k, v in fetches.items():
                flat[k] = self.run(v)
            return flat
        if hasattr(fetches, '_fields'):  # namedtuple
            vals = self._evaluate_list(list(fetches))
            return fetches.__class__(*vals)
        vals = self._evaluate_list(list(fetches))
        return type(fetches)(vals)


def py_func(func, inp, dtypes):
    assert inp == [], 'tf_stub py_func supports no graph inputs'
    node = _PyFuncNode(func, dtypes)
    return [Tensor(node, i, dt) for i, dt in enumerate(dtypes)]


class _EagerValue(object):
    """What iterating a Dataset yields per field (models EagerTensor)."""

    def __init__(self, value, dtype):
        self._value = value
        self.dtype = dtype
        self._shape = None

    def set_shape(self, shape):
        self._shape = tuple(shape)
        _check_value(self._value, self.dtype, self._shape, 'set_shape')

    def numpy(self):
        return self._value


def _apply_element_fn(fn, element):
    """tf.data calls map/flat_map fns with plain tuples unpacked but
    namedtuples passed whole (nest semantics)."""
    if isinstance(element, tuple) and not hasattr(element, '_fields'):
        return fn(*element)
    return fn(element)


def _as_eager(v):
    if isinstance(v, _EagerValue):
        return v
    arr = np.asarray(v)
    dt = string if arr.dtype.kind in 'SUO' else as_dtype(arr.dtype)
    return _EagerValue(arr if dt is not string else v, dt)


AUTOTUNE = -1


class Dataset(object):
    """Element-iterator dataset supporting the ops the adapters use:
    from_generator, from_tensor_slices, map, flat_map, batch, prefetch."""

    def __init__(self, factory):
        self._factory = factory  # callable -> iterator of elements

    @staticmethod
    def from_generator(generator, output_types):
        dtypes = tuple(output_types)

        def factory():
            for vals in generator():
                if not isinstance(vals, tuple):
                    vals = (vals,)
                yield tuple(
                    _EagerValue(_check_value(v, dt, None, 'dataset'), dt)
                    for v, dt in zip(vals, dtypes))
        return Dataset(factory)

    @staticmethod
    def from_tensor_slices(element):
        def factory():
            if hasattr(element, '_fields'):
                fields = [_as_eager(v) for v in element]
                n = len(np.asarray(fields[0].numpy()))
                for i in range(n):
                    yield element.__class__(*[
                        _as_eager(np.asarray(f.numpy())[i])
                        for f in fields])
            else:
                parts = element if isinstance(element, tuple) else (element,)
                fields = [_as_eager(v) for v in parts]
                n = len(np.asarray(fields[0].numpy()))
                for i in range(n):
                    yield tuple(_as_eager(np.asarray(f.numpy())[i])
                                for f in fields)
        return Dataset(factory)

    def map(self, fn):
        def factory():
            for e in self._factory():
                yield _apply_element_fn(fn, e)
        return Dataset(factory)

    def flat_map(self, fn):
        def factory():
            for e in self._factory():
                sub = _apply_element_fn(fn, e)
                for x in sub:
                    yield x
        return Dataset(factory)

    def batch(self, batch_size, drop_remainder=False):
        def factory():
            buf = []
            for e in self._factory():
                buf.append(e)
                if len(buf) == batch_size:
                    yield self._stack(buf)
                    buf = []
            if buf and not drop_remainder:
                yield self._stack(buf)
        return Dataset(factory)

    @staticmethod
    def _stack(elements):
        first = elements[0]

        def stack_field(i_or_name):
            vals = [np.asarray((e[i_or_name] if isinstance(i_or_name, int)
                                else getattr(e, i_or_name)).numpy())
                    for e in elements]
            try:
                return _as_eager(np.stack(vals))
            except ValueError:
                arr = np.empty(len(vals), dtype=object)
                arr[:] = vals
                return _as_eager(arr)
        if hasattr(first, '_fields'):
            return first.__class__(*[stack_field(i)
                                     for i in range(len(first))])
        return tuple(stack_field(i) for i in range(len(first)))

    def prefetch(self, n):
        return self

    def __iter__(self):
        return iter(self._factory())


def build_module():
    """Assemble a module object shaped like the tensorflow package."""
    tf = types.ModuleType('tensorflow')
    for name in ('string', 'int8', 'int16', 'int32', 'int64', 'uint8',
                 'float32', 'float64'):
        setattr(tf, name, globals()[name])
    tf.bool = bool_
    tf.as_dtype = as_dtype
    tf.identity = identity
    tf.Tensor = Tensor

    tf.queue = types.ModuleType('tensorflow.queue')
    tf.queue.RandomShuffleQueue = RandomShuffleQueue

    tf.compat = types.ModuleType('tensorflow.compat')
    tf.compat.v1 = types.ModuleType('tensorflow.compat.v1')
    tf.compat.v1.py_func = py_func
    tf.compat.v1.Session = Session
    tf.compat.v1.train = types.ModuleType('tensorflow.compat.v1.train')
    tf.compat.v1.train.QueueRunner = QueueRunner
    tf.compat.v1.train.add_queue_runner = _add_queue_runner

    tf.data = types.ModuleType('tensorflow.data')
    tf.data.Dataset = Dataset
    tf.data.AUTOTUNE = AUTOTUNE
    return tf
