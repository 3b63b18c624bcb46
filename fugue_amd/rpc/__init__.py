from fugue_amd.rpc.base import (
    EmptyRPCHandler,
    NativeRPCClient,
    NativeRPCServer,
    RPCClient,
    RPCFunc,
    RPCHandler,
    RPCServer,
    make_rpc_server,
    to_rpc_handler,
)
