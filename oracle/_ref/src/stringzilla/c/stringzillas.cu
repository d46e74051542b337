/**
 *  @file       stringzillas.cu
 *  @brief      StringZillas library for parallel string operations using CUDA backends.
 *  @author     Ash Vardanian
 *  @date       March 23, 2025
 */
#include "stringzillas.cuh"