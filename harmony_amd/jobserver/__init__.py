"""Job server: long-running master accepting concurrent PS job submissions.

Reference: jobserver/src/.../jobserver — JobServerClient/CommandSender
(socket localhost:7008), JobServerDriver (SUBMIT/SHUTDOWN handling,
ResourcePool, JobScheduler SPI, JobDispatcher threads).

MI355X shape: the "driver" is rank 0 of the long-running N-GPU executor
group (one process per GPU, launched once by start_jobserver.sh). Rank 0
listens on the command socket; accepted jobs are fanned out through the
control store; every rank runs a dispatcher loop that joins each job
collectively (forming the job's process subgroup) — jobs co-locate on the
same GPUs via HIP streams + the global task-unit sequencer.
"""
