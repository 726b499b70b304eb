from rayfed_amd.proxy.grpc.grpc_proxy import GrpcReceiverProxy, GrpcSenderProxy

__all__ = ["GrpcSenderProxy", "GrpcReceiverProxy"]
