"""embedding_lookup with mod/div partition strategies (reference
python/ops/embedding_ops.py:44 — sharded variables across PS tasks map to
sharded lookups here)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import array_ops, math_ops


def embedding_lookup(params, ids, partition_strategy='mod', name=None,
                     validate_indices=True, max_norm=None):
    if not isinstance(params, (list, tuple)):
        params = [params]
    params = [p._as_graph_element() if hasattr(p, '_as_graph_element') else p
              for p in params]
    ids = convert_to_tensor(ids)
    np_ = len(params)
    if np_ == 1:
        return array_ops.gather(params[0], ids, name=name)
    g = ops.get_default_graph()
    with g.name_scope(name or 'embedding_lookup'):
        flat_ids = array_ops.reshape(ids, [-1])
        if partition_strategy == 'mod':
            p_assignments = math_ops.floormod(flat_ids, np_)
            new_ids = math_ops.floordiv(flat_ids, np_)
        elif partition_strategy == 'div':
            # even-ish split over the total rows of all partitions
            total = sum(p._shape[0] for p in params)
            per = (total + np_ - 1) // np_
            p_assignments = math_ops.floordiv(flat_ids, per)
            new_ids = math_ops.floormod(flat_ids, per)
        else:
            raise ValueError('unknown partition_strategy %r'
                             % partition_strategy)
        # Gather from each partition, then select rows by assignment.
        parts = []
        for i, p in enumerate(params):
            mask = math_ops.cast(
                math_ops.equal(p_assignments,
                               ops.constant(i, dtype=p_assignments.dtype)),
                new_ids.dtype)
            safe_ids = new_ids * mask  # out-of-partition rows read row 0
            gathered = array_ops.gather(p, safe_ids)
            fmask = math_ops.cast(mask, gathered.dtype)
            parts.append(gathered * array_ops.reshape(fmask, [-1, 1]))
        out = parts[0]
        for p in parts[1:]:
            out = out + p
        if ids._shape is not None and len(ids._shape) > 1:
            emb_dim = params[0]._shape[1]
            out = array_ops.reshape(out, list(ids._shape) + [emb_dim])
        return out


def embedding_lookup_sparse(params, sp_ids, sp_weights, combiner='mean',
                            name=None, partition_strategy='mod'):
    raise NotImplementedError(
        'embedding_lookup_sparse requires SparseTensor support')
