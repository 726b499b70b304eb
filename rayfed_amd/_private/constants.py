"""Internal constants.

Parity: /root/reference/fed/_private/constants.py:20-44 (KV keys, log format,
default job name, proxy service default names).
"""

KEY_OF_CLUSTER_CONFIG = "CLUSTER_CONFIG"
KEY_OF_JOB_CONFIG = "JOB_CONFIG"

KEY_OF_CLUSTER_ADDRESSES = "CLUSTER_ADDRESSES"
KEY_OF_CURRENT_PARTY_NAME = "CURRENT_PARTY_NAME"
KEY_OF_TLS_CONFIG = "TLS_CONFIG"
KEY_OF_CROSS_SILO_COMM_CONFIG_DICT = "CROSS_SILO_COMM_CONFIG_DICT"

RAYFED_LOG_FMT = (
    "%(asctime)s.%(msecs)03d %(levelname)s %(filename)s:%(lineno)s"
    " [%(party)s] -- [%(jobname)s] %(message)s"
)
RAYFED_DATE_FMT = "%Y-%m-%d %H:%M:%S"

RAYFED_DEFAULT_JOB_NAME = "Anonymous_job"

RAYFED_DEFAULT_SENDER_PROXY_NAME = "SenderProxy"
RAYFED_DEFAULT_RECEIVER_PROXY_NAME = "ReceiverProxy"
RAYFED_DEFAULT_SENDER_RECEIVER_PROXY_NAME = "SenderReceiverProxy"

# Seq id used by the init-time readiness barrier (ping_others).
PING_SEQ_ID = "ping"
