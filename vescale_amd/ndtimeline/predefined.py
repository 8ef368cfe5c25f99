"""Predefined metric names (parity: legacy/vescale/ndtimeline/predefined.py:18-31)."""

FORWARD_COMPUTE = "forward-compute"
BACKWARD_COMPUTE = "backward-compute"
UNSHARD_AG = "unshard-all-gather"
GRAD_RS = "grad-reduce-scatter"
GRAD_AR = "grad-all-reduce"
PARAM_AG = "param-all-gather"
OPTIMIZER_STEP = "optimizer-step"
RECV_FORWARD = "recv-forward"
RECV_BACKWARD = "recv-backward"
SEND_FORWARD = "send-forward"
SEND_BACKWARD = "send-backward"
SEND_FORWARD_RECV_BACKWARD = "send-forward-recv-backward"
SEND_BACKWARD_RECV_FORWARD = "send-backward-recv-forward"
CROSS_MESH_RECV = "cross-mesh-recv"
CROSS_MESH_SEND = "cross-mesh-send"

ALL = [v for k, v in list(globals().items()) if k.isupper() and isinstance(v, str)]
