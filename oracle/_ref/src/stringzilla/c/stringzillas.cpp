/**
 *  @file       stringzillas.cpp
 *  @brief      StringZillas library for parallel string operations using CPU backends.
 *  @author     Ash Vardanian
 *  @date       March 23, 2025
 */
#include "stringzillas.cuh"