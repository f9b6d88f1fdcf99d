/* shim forward + version macros (pretend to be a modern kernel) */
#ifndef ROCNR_SHIM_LINUX_VERSION_H_
#define ROCNR_SHIM_LINUX_VERSION_H_
#define KERNEL_VERSION(a, b, c) (((a) << 16) + ((b) << 8) + (c))
#define LINUX_VERSION_CODE KERNEL_VERSION(6, 8, 0)
#include "../rocnr_shim_all.h"
#endif
