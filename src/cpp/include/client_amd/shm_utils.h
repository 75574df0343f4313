// POSIX system shared-memory helpers (reference:
// src/c++/library/shm_utils.{h,cc}:39-106 — same function names).
#pragma once

#include <string>

#include "client_amd/common.h"

namespace client_amd {

// shm_open + ftruncate; returns fd.
Error CreateSharedMemoryRegion(
    const std::string& shm_key, size_t byte_size, int* shm_fd);

// mmap the region; returns base pointer at offset.
Error MapSharedMemory(
    int shm_fd, size_t offset, size_t byte_size, void** shm_addr);

Error CloseSharedMemory(int shm_fd);

Error UnlinkSharedMemoryRegion(const std::string& shm_key);

Error UnmapSharedMemory(void* shm_addr, size_t byte_size);

}  // namespace client_amd
