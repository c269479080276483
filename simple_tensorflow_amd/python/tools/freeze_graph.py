"""freeze_graph: GraphDef + checkpoint -> single self-contained GraphDef with
variables folded into constants (capability analog of reference
python/tools/freeze_graph.py)."""
import argparse

from simple_tensorflow_amd.python.framework import graph_util, ops


def freeze_graph_from_session(sess, output_node_names):
    gd = sess.graph.as_graph_def() if hasattr(sess, 'graph') else \
        ops.get_default_graph().as_graph_def()
    return graph_util.convert_variables_to_constants(
        sess, gd, output_node_names)


def freeze_graph(input_graph, input_checkpoint, output_node_names,
                 output_graph, input_binary=True, input_saver=None,
                 restore_op_name=None, filename_tensor_name=None,
                 clear_devices=True, initializer_nodes=None):
    """Loads input_graph (GraphDef bytes file), restores input_checkpoint,
    writes the frozen GraphDef to output_graph."""
    import simple_tensorflow_amd as tf
    from simple_tensorflow_amd.python.training import saver as saver_mod

    with open(input_graph, 'rb') as f:
        gd = f.read()
    names = [n.strip() for n in output_node_names.split(',') if n.strip()]
    g = ops.Graph()
    with g.as_default():
        tf.import_graph_def(gd, name='')
        with tf.Session() as sess:
            meta = input_checkpoint + '.meta'
            import os
            if os.path.exists(meta):
                # the meta graph duplicates the imported graph; only its
                # saver wiring is needed, so build one from the raw graph
                pass
            # No Variable objects exist after a raw import: build the saver
            # over the graph's VariableV2 ref tensors directly.
            var_list = {
                op.name: op.outputs[0]
                for op in g._node_list if op.type in ('Variable',
                                                      'VariableV2')}
            sv = saver_mod.Saver(var_list=var_list)
            sv.restore(sess, input_checkpoint)
            if initializer_nodes:
                for n in initializer_nodes.split(','):
                    sess.run(g.get_operation_by_name(n.strip()))
            frozen = graph_util.convert_variables_to_constants(
                sess, g.as_graph_def(), names)
    if clear_devices:
        nodes = graph_util._as_nodes(frozen)
        for n in nodes:
            n['device'] = ''
        frozen = graph_util._serialize(nodes)
    with open(output_graph, 'wb') as f:
        f.write(frozen)
    return frozen


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--input_graph', required=True)
    p.add_argument('--input_checkpoint', required=True)
    p.add_argument('--output_graph', required=True)
    p.add_argument('--output_node_names', required=True)
    p.add_argument('--initializer_nodes', default='')
    a = p.parse_args()
    freeze_graph(a.input_graph, a.input_checkpoint, a.output_node_names,
                 a.output_graph, initializer_nodes=a.initializer_nodes)


if __name__ == '__main__':
    main()
