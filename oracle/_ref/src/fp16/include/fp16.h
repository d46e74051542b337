#pragma once
#ifndef FP16_H
#define FP16_H

#include <fp16/fp16.h>

#endif /* FP16_H */
