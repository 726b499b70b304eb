from rayfed_amd.proxy.tcp.tcp_proxy import TcpReceiverProxy, TcpSenderProxy

__all__ = ["TcpSenderProxy", "TcpReceiverProxy"]
