/* Single-rank MPI shim implementation (see mpi.h). */

#include "mpi.h"

#include <stdlib.h>

static sbgmpi_type t_int = {sizeof(int)};
static sbgmpi_type t_u64 = {8};
static sbgmpi_type t_u16 = {2};
static sbgmpi_type t_u8 = {1};
static sbgmpi_type t_bool = {1};

MPI_Datatype MPI_INT = &t_int;
MPI_Datatype MPI_UINT64_T = &t_u64;
MPI_Datatype MPI_UINT16_T = &t_u16;
MPI_Datatype MPI_UINT8_T = &t_u8;
MPI_Datatype MPI_C_BOOL = &t_bool;

int MPI_Init(int* argc, char*** argv) { (void)argc; (void)argv; return MPI_SUCCESS; }
int MPI_Finalize(void) { return MPI_SUCCESS; }
int MPI_Comm_rank(MPI_Comm comm, int* rank) { (void)comm; *rank = 0; return MPI_SUCCESS; }
int MPI_Comm_size(MPI_Comm comm, int* size) { (void)comm; *size = 1; return MPI_SUCCESS; }
int MPI_Barrier(MPI_Comm comm) { (void)comm; return MPI_SUCCESS; }

int MPI_Bcast(void* buf, int count, MPI_Datatype type, int root, MPI_Comm comm) {
  /* Root is always self at world size 1. */
  (void)buf; (void)count; (void)type; (void)root; (void)comm;
  return MPI_SUCCESS;
}

/* Point-to-point: at world size 1 the reference only ever posts requests
 * that it later cancels (the early-exit Irecv) or sends to itself that it
 * never matches (it uses quit_msg directly when rank == 0). Requests are
 * represented as "pending" (1) and never complete. */
int MPI_Isend(const void* buf, int count, MPI_Datatype type, int dest, int tag,
              MPI_Comm comm, MPI_Request* req) {
  (void)buf; (void)count; (void)type; (void)dest; (void)tag; (void)comm;
  *req = 1;
  return MPI_SUCCESS;
}
int MPI_Irecv(void* buf, int count, MPI_Datatype type, int source, int tag,
              MPI_Comm comm, MPI_Request* req) {
  (void)buf; (void)count; (void)type; (void)source; (void)tag; (void)comm;
  *req = 1;
  return MPI_SUCCESS;
}
int MPI_Test(MPI_Request* req, int* flag, MPI_Status* status) {
  (void)status;
  *flag = *req == MPI_REQUEST_NULL ? 1 : 0;
  return MPI_SUCCESS;
}
int MPI_Cancel(MPI_Request* req) { *req = MPI_REQUEST_NULL; return MPI_SUCCESS; }
int MPI_Wait(MPI_Request* req, MPI_Status* status) {
  (void)status;
  *req = MPI_REQUEST_NULL;
  return MPI_SUCCESS;
}
int MPI_Waitall(int count, MPI_Request* reqs, MPI_Status* statuses) {
  (void)statuses;
  for (int i = 0; i < count; i++) reqs[i] = MPI_REQUEST_NULL;
  return MPI_SUCCESS;
}
int MPI_Iprobe(int source, int tag, MPI_Comm comm, int* flag, MPI_Status* status) {
  (void)source; (void)tag; (void)comm; (void)status;
  *flag = 0;
  return MPI_SUCCESS;
}
int MPI_Recv(void* buf, int count, MPI_Datatype type, int source, int tag,
             MPI_Comm comm, MPI_Status* status) {
  (void)buf; (void)count; (void)type; (void)source; (void)tag; (void)comm;
  (void)status;
  return MPI_SUCCESS;
}

int MPI_Allgather(const void* sendbuf, int sendcount, MPI_Datatype sendtype,
                  void* recvbuf, int recvcount, MPI_Datatype recvtype,
                  MPI_Comm comm) {
  (void)recvcount; (void)recvtype; (void)comm;
  memcpy(recvbuf, sendbuf, (size_t)sendcount * sendtype->extent);
  return MPI_SUCCESS;
}
int MPI_Allgatherv(const void* sendbuf, int sendcount, MPI_Datatype sendtype,
                   void* recvbuf, const int* recvcounts, const int* displs,
                   MPI_Datatype recvtype, MPI_Comm comm) {
  (void)recvcounts; (void)recvtype; (void)comm;
  memcpy((char*)recvbuf + displs[0] * sendtype->extent, sendbuf,
         (size_t)sendcount * sendtype->extent);
  return MPI_SUCCESS;
}

int MPI_Type_create_struct(int count, const int* block_lengths,
                           const MPI_Aint* displacements,
                           const MPI_Datatype* types, MPI_Datatype* newtype) {
  /* Only the extent matters for the copies above; approximate it as the
   * end of the last block. */
  size_t extent = 0;
  for (int i = 0; i < count; i++) {
    size_t end = (size_t)displacements[i] +
                 (size_t)block_lengths[i] * types[i]->extent;
    if (end > extent) extent = end;
  }
  sbgmpi_type* t = (sbgmpi_type*)malloc(sizeof(sbgmpi_type));
  t->extent = extent;
  *newtype = t;
  return MPI_SUCCESS;
}
int MPI_Type_create_resized(MPI_Datatype oldtype, MPI_Aint lb, MPI_Aint extent,
                            MPI_Datatype* newtype) {
  (void)lb;
  sbgmpi_type* t = (sbgmpi_type*)malloc(sizeof(sbgmpi_type));
  t->extent = (size_t)extent;
  (void)oldtype;
  *newtype = t;
  return MPI_SUCCESS;
}
int MPI_Type_commit(MPI_Datatype* type) { (void)type; return MPI_SUCCESS; }
