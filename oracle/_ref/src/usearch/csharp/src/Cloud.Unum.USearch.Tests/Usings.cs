global using Xunit;
