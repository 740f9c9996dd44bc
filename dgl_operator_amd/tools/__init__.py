"""Workflow-plane CLIs (the reference's python/dglrun toolbox):

python -m dgl_operator_amd.tools.dglrun      # 5-phase GNN job driver
python -m dgl_operator_amd.tools.dglkerun    # DGL-KE job driver
python -m dgl_operator_amd.tools.launch      # exec/copy/train fabric
python -m dgl_operator_amd.tools.dispatch    # ship partitions to workers
python -m dgl_operator_amd.tools.hostfile    # hostfile -> ipconfig revision
python -m dgl_operator_amd.tools.export_ke   # merge KGE shards to .npy
"""
