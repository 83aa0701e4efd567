"""trtlab_amd.rpc — asynchronous gRPC microservice framework.

MI355X-native redesign of trtlab/nvrpc: Server /
AsyncService / unary+streaming+batching lifecycles / executors / clients,
built on grpc.aio completion-queue machinery. Messages are defined with
dynamic protobuf descriptors (no protoc needed offline) in rpc.proto.
"""
from trtlab_amd.rpc.proto import (EchoRequest, EchoResponse, InferRequest,  # noqa: F401
                                  InferResponse, HealthRequest,
                                  HealthResponse, NamedTensor)
from trtlab_amd.rpc.server import (AsyncService, BatchingService, Server,  # noqa: F401
                                   StreamingService, UnaryService)
from trtlab_amd.rpc.client import (AsyncClient, ShmInput, SyncClient,  # noqa: F401
                                   siege)
from trtlab_amd.rpc.remote import (RemoteInferenceManager,  # noqa: F401
                                   RemoteInferRunner)
