/* Synthetic config.h for compiling the reference EC/CRC sources directly
 * from /root/reference (see oracle/Makefile).  Values per the reference's
 * CMakeLists.txt:93-94 (LIZARDFS_BLOCKS_IN_CHUNK=1024, BLOCK_SIZE=65536). */
#pragma once
#define MFSBLOCKSIZE 65536
#define MFSBLOCKSINCHUNK 1024
#define ENABLE_CRC 1
#define LIZARDFS_HAVE_STD_TO_STRING 1
#define LIZARDFS_HAVE_STD_STOULL 1
