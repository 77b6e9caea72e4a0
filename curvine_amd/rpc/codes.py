"""RPC operation codes.

Same numeric surface as the reference's `RpcCode`
(/root/reference/crates/common/curvine-fs-api/src/rpc_code.rs:20-91) so the
two systems' wire logs are comparable.  Codes ≥100 are curvine_amd additions
for the MI355X data plane (device short-circuit, RCCL group setup).
"""
from __future__ import annotations

from enum import IntEnum


class RpcCode(IntEnum):
    Undefined = 0
    Heartbeat = 1

    # filesystem API
    Mkdir = 2
    Delete = 3
    CreateFile = 4
    OpenFile = 5
    AppendFile = 6
    FileStatus = 7
    ListStatus = 8
    Exists = 9
    Rename = 10
    AddBlock = 11
    CompleteFile = 12
    GetBlockLocations = 13
    GetFilesystemInfo = 14
    SetAttr = 15
    Symlink = 16
    Link = 17
    ResizeFile = 18
    AssignWorker = 19
    GetLock = 20
    SetLock = 21
    ListLock = 22
    CreateFilesBatch = 23
    AddBlocksBatch = 24
    CompleteFilesBatch = 25
    Free = 26
    ListOptions = 27
    GetMetadataSnapshotPage = 28
    GetMetadataDeltaPage = 29

    # mount manager
    Mount = 30
    UnMount = 31
    UpdateMount = 32
    GetMountTable = 33
    GetMountInfo = 34

    # load jobs
    SubmitJob = 35
    GetJobStatus = 36
    CancelJob = 37
    ReportTask = 38
    SubmitTask = 39

    # worker <-> master
    WorkerHeartbeat = 40
    WorkerBlockReport = 41

    # replication
    SubmitBlockReplicationJob = 42
    ReportBlockReplicationResult = 43
    RequestReplacementWorker = 44
    ReportUnderReplicatedBlocks = 45

    # transfer service
    SubmitTransfer = 46
    GetTransferStatus = 47
    CancelTransfer = 48
    ReportTransferTask = 49
    QueryTransferTask = 50
    WatchTransfer = 51
    ListTransfers = 52
    ListTransferTenants = 53
    RetryTransfer = 54

    MetricsReport = 60
    DecommissionWorker = 61   # curvine_amd admin extension

    # raft journal (curvine_amd: raft speaks the same RPC framing)
    RaftVote = 70
    RaftAppendEntries = 71
    RaftInstallSnapshot = 72
    RaftTransferLeader = 73

    # block interface (worker data plane)
    WriteBlock = 80
    ReadBlock = 81
    WriteBlocksBatch = 82
    WriteCommitsBatch = 83

    # ---- curvine_amd additions ----
    ShortCircuitInfo = 100     # disclose local block location (path or HBM extent)
    UnpinBlock = 101           # release a ShortCircuitInfo pin lease
    DeviceShortCircuit = 101   # hipIpc/dmabuf handle for cross-process GPU read
    RcclGroupSetup = 102       # establish an RCCL communicator for bulk distribution
