"""SavedModel builder/loader — byte-compatible wire layout with the
reference core/protobuf/saved_model.proto (saved_model_schema_version=1,
meta_graphs=2) and the reference directory layout
(python/saved_model/constants.py:40-49: saved_model.pb +
variables/variables.{index,data-*}).

Capability analog of reference python/saved_model/{builder_impl,
loader_impl,tag_constants,signature_constants,signature_def_utils,
utils_impl}.py."""
import os

from simple_tensorflow_amd.python.framework import ops
from simple_tensorflow_amd.python.framework import meta_graph as mg
from simple_tensorflow_amd.python.framework import pbreader
from simple_tensorflow_amd.python.framework.pbwire import (
    f_bytes, f_varint, tensor_shape_proto)

SAVED_MODEL_FILENAME_PB = 'saved_model.pb'
VARIABLES_DIRECTORY = 'variables'
VARIABLES_FILENAME = 'variables'
SAVED_MODEL_SCHEMA_VERSION = 1


class tag_constants(object):
    SERVING = 'serve'
    TRAINING = 'train'
    GPU = 'gpu'


class signature_constants(object):
    DEFAULT_SERVING_SIGNATURE_DEF_KEY = 'serving_default'
    PREDICT_INPUTS = 'inputs'
    PREDICT_METHOD_NAME = 'tensorflow/serving/predict'
    PREDICT_OUTPUTS = 'outputs'
    CLASSIFY_INPUTS = 'inputs'
    CLASSIFY_METHOD_NAME = 'tensorflow/serving/classify'
    CLASSIFY_OUTPUT_CLASSES = 'classes'
    CLASSIFY_OUTPUT_SCORES = 'scores'
    REGRESS_INPUTS = 'inputs'
    REGRESS_METHOD_NAME = 'tensorflow/serving/regress'
    REGRESS_OUTPUTS = 'outputs'


# ---- TensorInfo / SignatureDef wire helpers (meta_graph.proto: TensorInfo
# name=1, dtype=2, tensor_shape=3; SignatureDef inputs=1, outputs=2,
# method_name=3) ----

def tensor_info_bytes(tensor):
    out = f_bytes(1, tensor.name)
    out += f_varint(2, int(tensor.dtype))
    if tensor._shape is not None:
        out += f_bytes(3, tensor_shape_proto(list(tensor._shape)))
    return out


def build_tensor_info(tensor):
    """Returns a dict form of TensorInfo (reference utils_impl.py)."""
    return {'name': tensor.name, 'dtype': int(tensor.dtype),
            'shape': list(tensor._shape) if tensor._shape is not None
            else None}


def _tensor_info_bytes_from_dict(ti):
    out = f_bytes(1, ti['name'])
    out += f_varint(2, ti['dtype'])
    if ti.get('shape') is not None:
        out += f_bytes(3, tensor_shape_proto(ti['shape']))
    return out


def signature_def_bytes(inputs, outputs, method_name):
    body = b''
    for k, ti in sorted(inputs.items()):
        entry = f_bytes(1, k) + f_bytes(2, _tensor_info_bytes_from_dict(ti))
        body += f_bytes(1, entry)
    for k, ti in sorted(outputs.items()):
        entry = f_bytes(1, k) + f_bytes(2, _tensor_info_bytes_from_dict(ti))
        body += f_bytes(2, entry)
    body += f_bytes(3, method_name)
    return body


def predict_signature_def(inputs, outputs):
    """inputs/outputs: dict name -> Tensor. Returns the signature in the
    builder's dict form (reference signature_def_utils_impl.py)."""
    return {
        'inputs': {k: build_tensor_info(v) for k, v in inputs.items()},
        'outputs': {k: build_tensor_info(v) for k, v in outputs.items()},
        'method_name': signature_constants.PREDICT_METHOD_NAME,
    }


def _parse_tensor_info(data):
    ti = {'name': '', 'dtype': 0, 'shape': None}
    for f, w, v in pbreader._fields(data):
        if f == 1:
            ti['name'] = v.decode()
        elif f == 2:
            ti['dtype'] = v
        elif f == 3:
            ti['shape'] = pbreader.parse_tensor_shape(v)
    return ti


def _parse_signature_def(data):
    sig = {'inputs': {}, 'outputs': {}, 'method_name': ''}
    for f, w, v in pbreader._fields(data):
        if f in (1, 2):
            key, ti = None, None
            for f2, _, v2 in pbreader._fields(v):
                if f2 == 1:
                    key = v2.decode()
                elif f2 == 2:
                    ti = _parse_tensor_info(v2)
            if key is not None:
                sig['inputs' if f == 1 else 'outputs'][key] = ti
        elif f == 3:
            sig['method_name'] = v.decode()
    return sig


class SavedModelBuilder(object):
    """Writes export_dir/saved_model.pb + variables/ via the Saver
    (reference builder_impl.py SavedModelBuilder)."""

    def __init__(self, export_dir):
        self._export_dir = export_dir
        if os.path.exists(os.path.join(export_dir, SAVED_MODEL_FILENAME_PB)):
            raise AssertionError(
                'Export directory already contains a saved model: %s' %
                export_dir)
        os.makedirs(export_dir, exist_ok=True)
        self._meta_graphs = []
        self._has_variables = False

    def _meta_info_bytes(self, tags):
        body = f_bytes(1, 'v1')  # meta_graph_version
        for t in tags:
            body += f_bytes(4, t)  # MetaInfoDef.tags = 4
        return body

    def add_meta_graph_and_variables(self, sess, tags, signature_def_map=None,
                                     assets_collection=None,
                                     legacy_init_op=None,
                                     clear_devices=False, main_op=None):
        from simple_tensorflow_amd.python.training import saver as saver_mod
        var_dir = os.path.join(self._export_dir, VARIABLES_DIRECTORY)
        os.makedirs(var_dir, exist_ok=True)
        saver = saver_mod.Saver()
        saver.save(sess, os.path.join(var_dir, VARIABLES_FILENAME),
                   write_meta_graph=False)
        self._add_meta_graph(tags, signature_def_map, saver)
        self._has_variables = True

    def add_meta_graph(self, tags, signature_def_map=None, **kw):
        if not self._has_variables:
            raise AssertionError(
                'Graph containing the variables must be saved first: call '
                'add_meta_graph_and_variables()')
        self._add_meta_graph(tags, signature_def_map, None)

    def _add_meta_graph(self, tags, signature_def_map, saver):
        mgd = b''
        mgd += f_bytes(1, self._meta_info_bytes(tags))
        g = ops.get_default_graph()
        mgd += f_bytes(2, g.as_graph_def())
        if saver is not None:
            mgd += f_bytes(3, mg.saver_def_bytes(saver))
        for cname in g._collections:
            cd = mg._collection_def(cname, g.get_collection(cname))
            if cd is None:
                continue
            mgd += f_bytes(4, f_bytes(1, cname) + f_bytes(2, cd))
        for key, sig in sorted((signature_def_map or {}).items()):
            sd = signature_def_bytes(sig['inputs'], sig['outputs'],
                                     sig['method_name'])
            mgd += f_bytes(5, f_bytes(1, key) + f_bytes(2, sd))
        self._meta_graphs.append(mgd)

    def save(self, as_text=False):
        out = f_varint(1, SAVED_MODEL_SCHEMA_VERSION)
        for m in self._meta_graphs:
            out += f_bytes(2, m)
        path = os.path.join(self._export_dir, SAVED_MODEL_FILENAME_PB)
        with open(path, 'wb') as f:
            f.write(out)
        return path


def _parse_saved_model(path):
    with open(os.path.join(path, SAVED_MODEL_FILENAME_PB), 'rb') as f:
        data = f.read()
    metas = []
    for f_, w, v in pbreader._fields(data):
        if f_ == 2:
            metas.append(bytes(v))
    return metas


def _meta_graph_tags(mgd):
    for f, w, v in pbreader._fields(mgd):
        if f == 1:
            tags = []
            for f2, _, v2 in pbreader._fields(v):
                if f2 == 4:
                    tags.append(v2.decode())
            return tags
    return []


def _meta_graph_signatures(mgd):
    sigs = {}
    for f, w, v in pbreader._fields(mgd):
        if f == 5:
            key, sd = None, None
            for f2, _, v2 in pbreader._fields(v):
                if f2 == 1:
                    key = v2.decode()
                elif f2 == 2:
                    sd = _parse_signature_def(v2)
            if key is not None:
                sigs[key] = sd
    return sigs


class loader(object):
    @staticmethod
    def maybe_saved_model_directory(export_dir):
        return os.path.isfile(
            os.path.join(export_dir, SAVED_MODEL_FILENAME_PB))

    @staticmethod
    def load(sess, tags, export_dir):
        """Imports the tagged meta graph into sess's graph and restores
        variables; returns {'signatures': {...}, 'tags': [...]}."""
        metas = _parse_saved_model(export_dir)
        want = set(tags)
        for mgd in metas:
            if set(_meta_graph_tags(mgd)) == want:
                break
        else:
            raise RuntimeError(
                'MetaGraphDef with tags %r not found in SavedModel %s' %
                (tags, export_dir))
        saver = mg.import_meta_graph(mgd)
        if saver is not None:
            saver.restore(sess, os.path.join(
                export_dir, VARIABLES_DIRECTORY, VARIABLES_FILENAME))
        return {'tags': list(want),
                'signatures': _meta_graph_signatures(mgd)}


def maybe_saved_model_directory(export_dir):
    return loader.maybe_saved_model_directory(export_dir)


class builder(object):
    SavedModelBuilder = SavedModelBuilder
