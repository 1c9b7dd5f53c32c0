/* Minimal single-rank MPI shim — test harness only.
 *
 * Used to compile the UPSTREAM reference (mounted read-only at
 * /root/reference) as a single-process oracle for cross-validation tests
 * (tests/test_reference_cross.py): our XML must load in the reference and
 * vice versa, fingerprints/file names must agree, and gate counts must be
 * comparable. Implements exactly the MPI surface the reference uses, with
 * world size 1 (collectives become copies/no-ops; point-to-point requests
 * never complete and can be cancelled, matching the reference's
 * single-rank control flow).
 *
 * This is part of the new framework's TEST SUITE, not a runtime
 * dependency; the engine itself contains no MPI.
 */
#ifndef SBG_TEST_MPI_STUB_H_
#define SBG_TEST_MPI_STUB_H_

#include <stddef.h>
#include <string.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef long MPI_Aint;
typedef int MPI_Comm;
typedef int MPI_Request;
typedef struct { int MPI_SOURCE, MPI_TAG, MPI_ERROR; } MPI_Status;

typedef struct {
  size_t extent;   /* bytes per element */
} sbgmpi_type;
typedef sbgmpi_type* MPI_Datatype;

#define MPI_COMM_WORLD 0
#define MPI_SUCCESS 0
#define MPI_ANY_SOURCE (-1)
#define MPI_REQUEST_NULL 0
#define MPI_STATUS_IGNORE ((MPI_Status*)0)
#define MPI_STATUSES_IGNORE ((MPI_Status*)0)

extern MPI_Datatype MPI_INT;
extern MPI_Datatype MPI_UINT64_T;
extern MPI_Datatype MPI_UINT16_T;
extern MPI_Datatype MPI_UINT8_T;
extern MPI_Datatype MPI_C_BOOL;

int MPI_Init(int* argc, char*** argv);
int MPI_Finalize(void);
int MPI_Comm_rank(MPI_Comm comm, int* rank);
int MPI_Comm_size(MPI_Comm comm, int* size);
int MPI_Barrier(MPI_Comm comm);
int MPI_Bcast(void* buf, int count, MPI_Datatype type, int root, MPI_Comm comm);
int MPI_Isend(const void* buf, int count, MPI_Datatype type, int dest, int tag,
              MPI_Comm comm, MPI_Request* req);
int MPI_Irecv(void* buf, int count, MPI_Datatype type, int source, int tag,
              MPI_Comm comm, MPI_Request* req);
int MPI_Test(MPI_Request* req, int* flag, MPI_Status* status);
int MPI_Wait(MPI_Request* req, MPI_Status* status);
int MPI_Waitall(int count, MPI_Request* reqs, MPI_Status* statuses);
int MPI_Cancel(MPI_Request* req);
int MPI_Iprobe(int source, int tag, MPI_Comm comm, int* flag, MPI_Status* status);
int MPI_Recv(void* buf, int count, MPI_Datatype type, int source, int tag,
             MPI_Comm comm, MPI_Status* status);
int MPI_Allgather(const void* sendbuf, int sendcount, MPI_Datatype sendtype,
                  void* recvbuf, int recvcount, MPI_Datatype recvtype,
                  MPI_Comm comm);
int MPI_Allgatherv(const void* sendbuf, int sendcount, MPI_Datatype sendtype,
                   void* recvbuf, const int* recvcounts, const int* displs,
                   MPI_Datatype recvtype, MPI_Comm comm);
int MPI_Type_create_struct(int count, const int* block_lengths,
                           const MPI_Aint* displacements,
                           const MPI_Datatype* types, MPI_Datatype* newtype);
int MPI_Type_create_resized(MPI_Datatype oldtype, MPI_Aint lb, MPI_Aint extent,
                            MPI_Datatype* newtype);
int MPI_Type_commit(MPI_Datatype* type);

#ifdef __cplusplus
}
#endif

#endif /* SBG_TEST_MPI_STUB_H_ */
