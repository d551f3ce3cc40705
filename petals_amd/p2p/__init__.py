from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage, RpcStream
