"""libfuse-free FUSE server (raw /dev/fuse protocol).

Analog of the reference's curvine-fuse crate (27k LoC,
/root/reference/curvine-fuse/): see curvine_amd.fuse.abi (kernel structs),
.session (mount + channel loops), .ops (FileSystem trait implementation
over the curvine client).
"""
